// X11 XShm framebuffer capture (MIT-SHM XShmGetImage each frame).
// Behavior parity target: the reference engine's X11 capture path
// ("X11 (XShm/XFixes/XDamage)").
#include <cstdio>
#include <cstring>

#include <sys/ipc.h>
#include <sys/shm.h>

#include <X11/Xlib.h>
#include <X11/Xutil.h>

#include "capture.h"
#include "x11_compat.h"

namespace hipflux {
namespace {

class X11Source : public FrameSource {
 public:
  static std::unique_ptr<X11Source> open(const std::string& display, int x,
                                         int y, int w, int h) {
    Display* dpy = XOpenDisplay(display.empty() ? nullptr : display.c_str());
    if (!dpy) return nullptr;
    auto src = std::unique_ptr<X11Source>(new X11Source());
    src->dpy_ = dpy;
    src->root_ = DefaultRootWindow(dpy);
    XWindowAttributes attr{};
    XGetWindowAttributes(dpy, src->root_, &attr);
    src->x_ = x;
    src->y_ = y;
    src->w_ = w > 0 ? w : attr.width - x;
    src->h_ = h > 0 ? h : attr.height - y;
    src->use_shm_ = XShmQueryExtension(dpy);
    if (src->use_shm_) {
      src->img_ = XShmCreateImage(dpy, DefaultVisual(dpy, DefaultScreen(dpy)),
                                  attr.depth, ZPixmap, nullptr, &src->shm_,
                                  src->w_, src->h_);
      if (src->img_) {
        src->shm_.shmid =
            shmget(IPC_PRIVATE,
                   static_cast<size_t>(src->img_->bytes_per_line) * src->h_,
                   IPC_CREAT | 0600);
        if (src->shm_.shmid >= 0) {
          src->shm_.shmaddr = static_cast<char*>(shmat(src->shm_.shmid, nullptr, 0));
          src->img_->data = src->shm_.shmaddr;
          src->shm_.readOnly = False;
          XShmAttach(dpy, &src->shm_);
          XSync(dpy, False);
          shmctl(src->shm_.shmid, IPC_RMID, nullptr);  // auto-reap
        } else {
          XDestroyImage(src->img_);
          src->img_ = nullptr;
          src->use_shm_ = false;
        }
      } else {
        src->use_shm_ = false;
      }
    }
    return src;
  }

  ~X11Source() override {
    if (dpy_) {
      if (use_shm_ && img_) {
        XShmDetach(dpy_, &shm_);
        XDestroyImage(img_);
        if (shm_.shmaddr) shmdt(shm_.shmaddr);
      } else if (img_) {
        XDestroyImage(img_);
      }
      XCloseDisplay(dpy_);
    }
  }

  bool acquire(RawFrame& out) override {
    if (use_shm_) {
      if (!XShmGetImage(dpy_, root_, img_, x_, y_, AllPlanes)) return false;
    } else {
      if (img_) XDestroyImage(img_);
      img_ = XGetImage(dpy_, root_, x_, y_, w_, h_, AllPlanes, ZPixmap);
      if (!img_) return false;
    }
    out.data = reinterpret_cast<const uint8_t*>(img_->data);
    out.width = w_;
    out.height = h_;
    out.stride = img_->bytes_per_line;
    out.ts_ms = now_ms();
    return true;
  }

  int width() const override { return w_; }
  int height() const override { return h_; }

  bool cursor(CursorImage& out) override {
    if (!have_xfixes_) {
      int eb, er;
      have_xfixes_ = XFixesQueryExtension(dpy_, &eb, &er) ? 1 : -1;
    }
    if (have_xfixes_ < 0) return false;
    XFixesCursorImage* ci = XFixesGetCursorImage(dpy_);
    if (!ci) return false;
    out.width = ci->width;
    out.height = ci->height;
    out.hot_x = ci->xhot;
    out.hot_y = ci->yhot;
    out.x = ci->x - ci->xhot - x_;
    out.y = ci->y - ci->yhot - y_;
    out.serial = ci->cursor_serial;
    out.argb.resize(static_cast<size_t>(ci->width) * ci->height);
    for (size_t i = 0; i < out.argb.size(); ++i)
      out.argb[i] = static_cast<uint32_t>(ci->pixels[i]);
    XFree(ci);
    return true;
  }

 private:
  X11Source() = default;
  Display* dpy_ = nullptr;
  Window root_ = 0;
  XImage* img_ = nullptr;
  XShmSegmentInfo shm_{};
  bool use_shm_ = false;
  int have_xfixes_ = 0;
  int x_ = 0, y_ = 0, w_ = 0, h_ = 0;
};

}  // namespace

std::unique_ptr<FrameSource> make_x11_source(const std::string& display, int x,
                                             int y, int w, int h) {
  return X11Source::open(display, x, y, w, h);
}

}  // namespace hipflux
