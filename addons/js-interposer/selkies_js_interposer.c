/* selkies_amd joystick interposer (LD_PRELOAD).
 *
 * Redirects opens of /dev/input/jsN (joydev) AND /dev/input/eventM with
 * M = 1000+N (evdev) to the selkies gamepad unix sockets
 * (SELKIES_JS_SOCKET_PATH/selkies_jsN.sock) so containerized apps see
 * gamepads without kernel devices — the architecture surveyed from the
 * reference js-interposer (it interposes both node
 * families, joystick_interposer.c:627-963), re-implemented against OUR
 * socket protocol (selkies_amd/gamepad.py):
 *   on connect the server sends a JsConfig struct:
 *     magic "SJSG", u16 version, u16 vendor, u16 product, u16 num_btns,
 *     u16 num_axes, char name[128], u16 btn_map[64], u8 axes_map[16]
 *   followed by a stream of `struct js_event` records (joydev ABI).
 *   js fds pass read() straight through; evdev fds TRANSLATE each
 *   js_event into struct input_event (+EV_SYN) in read().
 *
 * Interposes: open, open64, openat, ioctl, close, access, read, and the
 * stat family (the nodes must look like character devices). epoll/poll
 * need no interposition: pad fds are real unix sockets.
 * Emulated joydev ioctls: JSIOCGVERSION, JSIOCGAXES, JSIOCGBUTTONS,
 * JSIOCGNAME, JSIOCGAXMAP, JSIOCGBTNMAP. Emulated evdev ioctls:
 * EVIOCGVERSION, EVIOCGID, EVIOCGNAME, EVIOCGPHYS/UNIQ/PROP,
 * EVIOCGBIT(0/EV_KEY/EV_ABS), EVIOCGABS, EVIOCGKEY/LED/SW, EVIOCGRAB,
 * EVIOCGEFFECTS.
 *
 * Build: gcc -shared -fPIC -o selkies_js_interposer.so \
 *            selkies_js_interposer.c -ldl
 */
#define _GNU_SOURCE
#include <dlfcn.h>
#include <errno.h>
#include <fcntl.h>
#include <linux/joystick.h>
#include <pthread.h>
#include <stdarg.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#define MAX_PADS 4

typedef struct {
  char magic[4];
  uint16_t version;
  uint16_t vendor;
  uint16_t product;
  uint16_t num_btns;
  uint16_t num_axes;
  char name[128];
  uint16_t btn_map[64];
  uint8_t axes_map[16];
} __attribute__((packed)) js_config_t;

typedef struct {
  int fd;                 /* socket fd doubling as the device fd */
  int is_evdev;           /* 1 = translate js_event -> input_event */
  js_config_t config;
  /* pending translated bytes not yet consumed by read() */
  unsigned char pending[64];
  int pending_len;
  int pending_off;
} pad_state_t;

static pad_state_t g_pads[64];
static int g_npads = 0;
static pthread_mutex_t g_lock = PTHREAD_MUTEX_INITIALIZER;

static int (*real_open)(const char*, int, ...) = NULL;
static int (*real_open64)(const char*, int, ...) = NULL;
static int (*real_openat)(int, const char*, int, ...) = NULL;
static int (*real_ioctl)(int, unsigned long, ...) = NULL;
static int (*real_close)(int) = NULL;
static int (*real_access)(const char*, int) = NULL;
static ssize_t (*real_read)(int, void*, size_t) = NULL;

static void init_real(void) {
  if (!real_open) real_open = dlsym(RTLD_NEXT, "open");
  if (!real_open64) real_open64 = dlsym(RTLD_NEXT, "open64");
  if (!real_openat) real_openat = dlsym(RTLD_NEXT, "openat");
  if (!real_ioctl) real_ioctl = dlsym(RTLD_NEXT, "ioctl");
  if (!real_close) real_close = dlsym(RTLD_NEXT, "close");
  if (!real_access) real_access = dlsym(RTLD_NEXT, "access");
  if (!real_read) real_read = dlsym(RTLD_NEXT, "read");
}

/* returns pad index for /dev/input/jsN (N < MAX_PADS), else -1 */
static int js_index(const char* path) {
  int n;
  if (!path) return -1;
  if (sscanf(path, "/dev/input/js%d", &n) == 1 && n >= 0 && n < MAX_PADS)
    return n;
  return -1;
}

/* pad index for /dev/input/event(1000+N), else -1 (the event-node
 * numbering our fake-udev advertises) */
static int ev_index(const char* path) {
  int n;
  if (!path) return -1;
  if (sscanf(path, "/dev/input/event%d", &n) == 1 && n >= 1000 &&
      n < 1000 + MAX_PADS)
    return n - 1000;
  return -1;
}

static ssize_t read_full(int fd, void* buf, size_t n) {
  size_t got = 0;
  init_real();
  while (got < n) {
    /* must bypass our own read() interposer (evdev fds would recurse) */
    ssize_t r = real_read(fd, (char*)buf + got, n - got);
    if (r <= 0) {
      if (r < 0 && (errno == EINTR)) continue;
      return -1;
    }
    got += (size_t)r;
  }
  return (ssize_t)got;
}

static int open_pad(int idx, int evdev) {
  const char* dir = getenv("SELKIES_JS_SOCKET_PATH");
  if (!dir) dir = "/tmp/selkies_js";
  char path[sizeof(((struct sockaddr_un*)0)->sun_path)];
  snprintf(path, sizeof(path), "%s/selkies_js%d.sock", dir, idx);

  int fd = socket(AF_UNIX, SOCK_STREAM, 0);
  if (fd < 0) return -1;
  struct sockaddr_un addr;
  memset(&addr, 0, sizeof(addr));
  addr.sun_family = AF_UNIX;
  strncpy(addr.sun_path, path, sizeof(addr.sun_path) - 1);
  if (connect(fd, (struct sockaddr*)&addr, sizeof(addr)) != 0) {
    real_close(fd);
    errno = ENOENT;
    return -1;
  }
  js_config_t cfg;
  if (read_full(fd, &cfg, sizeof(cfg)) != (ssize_t)sizeof(cfg) ||
      memcmp(cfg.magic, "SJSG", 4) != 0) {
    real_close(fd);
    errno = EIO;
    return -1;
  }
  pthread_mutex_lock(&g_lock);
  if (g_npads < (int)(sizeof(g_pads) / sizeof(g_pads[0]))) {
    memset(&g_pads[g_npads], 0, sizeof(pad_state_t));
    g_pads[g_npads].fd = fd;
    g_pads[g_npads].is_evdev = evdev;
    g_pads[g_npads].config = cfg;
    ++g_npads;
  }
  pthread_mutex_unlock(&g_lock);
  return fd;
}

static pad_state_t* find_pad(int fd) {
  pad_state_t* out = NULL;
  pthread_mutex_lock(&g_lock);
  for (int i = 0; i < g_npads; ++i)
    if (g_pads[i].fd == fd) {
      out = &g_pads[i];
      break;
    }
  pthread_mutex_unlock(&g_lock);
  return out;
}

static void drop_pad(int fd) {
  pthread_mutex_lock(&g_lock);
  for (int i = 0; i < g_npads; ++i)
    if (g_pads[i].fd == fd) {
      g_pads[i] = g_pads[g_npads - 1];
      --g_npads;
      break;
    }
  pthread_mutex_unlock(&g_lock);
}

int open(const char* path, int flags, ...) {
  init_real();
  int idx = js_index(path);
  if (idx >= 0) return open_pad(idx, 0);
  idx = ev_index(path);
  if (idx >= 0) return open_pad(idx, 1);
  va_list ap;
  va_start(ap, flags);
  mode_t mode = va_arg(ap, mode_t);
  va_end(ap);
  return real_open(path, flags, mode);
}

int open64(const char* path, int flags, ...) {
  init_real();
  int idx = js_index(path);
  if (idx >= 0) return open_pad(idx, 0);
  idx = ev_index(path);
  if (idx >= 0) return open_pad(idx, 1);
  va_list ap;
  va_start(ap, flags);
  mode_t mode = va_arg(ap, mode_t);
  va_end(ap);
  return real_open64 ? real_open64(path, flags, mode)
                     : real_open(path, flags, mode);
}

int openat(int dirfd, const char* path, int flags, ...) {
  init_real();
  int idx = js_index(path);
  if (idx >= 0) return open_pad(idx, 0);
  idx = ev_index(path);
  if (idx >= 0) return open_pad(idx, 1);
  va_list ap;
  va_start(ap, flags);
  mode_t mode = va_arg(ap, mode_t);
  va_end(ap);
  return real_openat(dirfd, path, flags, mode);
}

int access(const char* path, int mode) {
  init_real();
  if (js_index(path) >= 0 || ev_index(path) >= 0)
    return 0; /* pretend the device exists */
  return real_access(path, mode);
}

/* ---- evdev read translation: js_event (8 B) -> input_event(s) -------- */
struct fake_input_event {
  uint64_t tv_sec;
  uint64_t tv_usec;
  uint16_t type;
  uint16_t code;
  int32_t value;
} __attribute__((packed));

static void push_iev(pad_state_t* p, uint16_t type, uint16_t code,
                     int32_t value, uint32_t ms) {
  struct fake_input_event ev;
  ev.tv_sec = ms / 1000u;
  ev.tv_usec = (uint64_t)(ms % 1000u) * 1000u;
  ev.type = type;
  ev.code = code;
  ev.value = value;
  memcpy(p->pending + p->pending_len, &ev, sizeof(ev));
  p->pending_len += (int)sizeof(ev);
}

ssize_t read(int fd, void* buf, size_t count) {
  init_real();
  pad_state_t* pad = find_pad(fd);
  if (!pad || !pad->is_evdev) return real_read(fd, buf, count);

  for (;;) {
    if (pad->pending_len > pad->pending_off) {
      int avail = pad->pending_len - pad->pending_off;
      int n = (int)count < avail ? (int)count : avail;
      memcpy(buf, pad->pending + pad->pending_off, (size_t)n);
      pad->pending_off += n;
      if (pad->pending_off >= pad->pending_len)
        pad->pending_off = pad->pending_len = 0;
      return n;
    }
    /* refill: one js_event becomes device event + EV_SYN */
    struct js_event je;
    ssize_t r = read_full(fd, &je, sizeof(je));
    if (r != (ssize_t)sizeof(je)) return r < 0 ? -1 : 0;
    if (je.type & JS_EVENT_INIT) continue;  /* evdev has no INIT records */
    js_config_t* c = &pad->config;
    if (je.type & JS_EVENT_BUTTON)
      push_iev(pad, 1 /*EV_KEY*/, c->btn_map[je.number % 64], je.value,
               je.time);
    else if (je.type & JS_EVENT_AXIS)
      push_iev(pad, 3 /*EV_ABS*/, c->axes_map[je.number % 16], je.value,
               je.time);
    else
      continue;
    push_iev(pad, 0 /*EV_SYN*/, 0, 0, je.time);
  }
}

/* ---- stat family: the fake nodes must look like char devices --------- */
#include <sys/stat.h>
#include <sys/sysmacros.h>

static int fake_stat_common(const char* path, struct stat* st) {
  int jn = js_index(path), en = ev_index(path);
  if (jn < 0 && en < 0) return 1;
  memset(st, 0, sizeof(*st));
  st->st_mode = S_IFCHR | 0660;
  st->st_rdev = jn >= 0 ? makedev(13, jn) : makedev(13, 64 + en);
  st->st_nlink = 1;
  return 0;
}

static int (*real_stat)(const char*, struct stat*) = NULL;
static int (*real_lstat)(const char*, struct stat*) = NULL;
static int (*real_xstat)(int, const char*, struct stat*) = NULL;
static int (*real_lxstat)(int, const char*, struct stat*) = NULL;

int stat(const char* path, struct stat* st) {
  if (!fake_stat_common(path, st)) return 0;
  if (!real_stat) real_stat = dlsym(RTLD_NEXT, "stat");
  if (real_stat) return real_stat(path, st);
  if (!real_xstat) real_xstat = dlsym(RTLD_NEXT, "__xstat");
  return real_xstat ? real_xstat(1, path, st) : -1;
}

int lstat(const char* path, struct stat* st) {
  if (!fake_stat_common(path, st)) return 0;
  if (!real_lstat) real_lstat = dlsym(RTLD_NEXT, "lstat");
  if (real_lstat) return real_lstat(path, st);
  if (!real_lxstat) real_lxstat = dlsym(RTLD_NEXT, "__lxstat");
  return real_lxstat ? real_lxstat(1, path, st) : -1;
}

int __xstat(int ver, const char* path, struct stat* st) {
  if (!fake_stat_common(path, st)) return 0;
  if (!real_xstat) real_xstat = dlsym(RTLD_NEXT, "__xstat");
  return real_xstat ? real_xstat(ver, path, st) : -1;
}

int __lxstat(int ver, const char* path, struct stat* st) {
  if (!fake_stat_common(path, st)) return 0;
  if (!real_lxstat) real_lxstat = dlsym(RTLD_NEXT, "__lxstat");
  return real_lxstat ? real_lxstat(ver, path, st) : -1;
}

/* modern glibc routes stat()/os.stat through fstatat/statx */
static int (*real_fstatat)(int, const char*, struct stat*, int) = NULL;
int fstatat(int dirfd, const char* path, struct stat* st, int flags) {
  if (!fake_stat_common(path, st)) return 0;
  if (!real_fstatat) real_fstatat = dlsym(RTLD_NEXT, "fstatat");
  return real_fstatat ? real_fstatat(dirfd, path, st, flags) : -1;
}

static int (*real_fxstatat)(int, int, const char*, struct stat*, int) = NULL;
int __fxstatat(int ver, int dirfd, const char* path, struct stat* st,
               int flags) {
  if (!fake_stat_common(path, st)) return 0;
  if (!real_fxstatat) real_fxstatat = dlsym(RTLD_NEXT, "__fxstatat");
  return real_fxstatat ? real_fxstatat(ver, dirfd, path, st, flags) : -1;
}

/* LFS-compiled callers (CPython among them) reference the 64 variants */
static int (*real_stat64)(const char*, void*) = NULL;
int stat64(const char* path, struct stat64* st) {
  if (!fake_stat_common(path, (struct stat*)st)) return 0;
  if (!real_stat64) real_stat64 = dlsym(RTLD_NEXT, "stat64");
  return real_stat64 ? real_stat64(path, st) : -1;
}

static int (*real_lstat64)(const char*, void*) = NULL;
int lstat64(const char* path, struct stat64* st) {
  if (!fake_stat_common(path, (struct stat*)st)) return 0;
  if (!real_lstat64) real_lstat64 = dlsym(RTLD_NEXT, "lstat64");
  return real_lstat64 ? real_lstat64(path, st) : -1;
}

static int (*real_fstatat64)(int, const char*, void*, int) = NULL;
int fstatat64(int dirfd, const char* path, struct stat64* st, int flags) {
  if (!fake_stat_common(path, (struct stat*)st)) return 0;
  if (!real_fstatat64) real_fstatat64 = dlsym(RTLD_NEXT, "fstatat64");
  return real_fstatat64 ? real_fstatat64(dirfd, path, st, flags) : -1;
}

static int (*real_xstat64)(int, const char*, void*) = NULL;
int __xstat64(int ver, const char* path, void* st) {
  if (!fake_stat_common(path, (struct stat*)st)) return 0;
  if (!real_xstat64) real_xstat64 = dlsym(RTLD_NEXT, "__xstat64");
  return real_xstat64 ? real_xstat64(ver, path, st) : -1;
}

static int (*real_lxstat64)(int, const char*, void*) = NULL;
int __lxstat64(int ver, const char* path, void* st) {
  if (!fake_stat_common(path, (struct stat*)st)) return 0;
  if (!real_lxstat64) real_lxstat64 = dlsym(RTLD_NEXT, "__lxstat64");
  return real_lxstat64 ? real_lxstat64(ver, path, st) : -1;
}

static int (*real_fxstatat64)(int, int, const char*, void*, int) = NULL;
int __fxstatat64(int ver, int dirfd, const char* path, void* st,
                 int flags) {
  if (!fake_stat_common(path, (struct stat*)st)) return 0;
  if (!real_fxstatat64) real_fxstatat64 = dlsym(RTLD_NEXT, "__fxstatat64");
  return real_fxstatat64 ? real_fxstatat64(ver, dirfd, path, st, flags)
                         : -1;
}

struct statx;
static int (*real_statx)(int, const char*, int, unsigned,
                         struct statx*) = NULL;
int statx(int dirfd, const char* path, int flags, unsigned mask,
          struct statx* sx) {
  int jn = js_index(path), en = ev_index(path);
  if (jn >= 0 || en >= 0) {
    /* fill via the plain-stat shape: statx layout offsets for mode and
     * rdev (glibc struct statx: stx_mode at offset 28 (u16),
     * stx_rdev_major at 128, stx_rdev_minor at 132) */
    unsigned char* b = (unsigned char*)sx;
    memset(b, 0, 256);
    *(uint16_t*)(b + 28) = S_IFCHR | 0660;       /* stx_mode */
    *(uint32_t*)(b + 128) = 13;                  /* stx_rdev_major */
    *(uint32_t*)(b + 132) = jn >= 0 ? (unsigned)jn : (unsigned)(64 + en);
    *(uint32_t*)(b + 0) = 0xFFFu;                /* stx_mask: basic stats */
    return 0;
  }
  if (!real_statx) real_statx = dlsym(RTLD_NEXT, "statx");
  if (!real_statx) { errno = ENOSYS; return -1; }
  return real_statx(dirfd, path, flags, mask, sx);
}

int ioctl(int fd, unsigned long request, ...) {
  init_real();
  va_list ap;
  va_start(ap, request);
  void* arg = va_arg(ap, void*);
  va_end(ap);

  pad_state_t* pad = find_pad(fd);
  if (!pad) return real_ioctl(fd, request, arg);

  js_config_t* c = &pad->config;
  unsigned dir = _IOC_DIR(request), type = _IOC_TYPE(request);
  unsigned nr = _IOC_NR(request), size = _IOC_SIZE(request);
  (void)dir;
  if (pad->is_evdev) {
    if (type != 'E') {
      errno = EINVAL;
      return -1;
    }
    if (nr == 0x01) {            /* EVIOCGVERSION */
      *(int*)arg = 0x010001;
      return 0;
    }
    if (nr == 0x02) {            /* EVIOCGID: bustype,vendor,product,ver */
      uint16_t* id = (uint16_t*)arg;
      id[0] = 3; /* BUS_USB */
      id[1] = c->vendor;
      id[2] = c->product;
      id[3] = c->version;
      return 0;
    }
    if (nr == 0x06) {            /* EVIOCGNAME(len) */
      size_t n = strnlen(c->name, sizeof(c->name));
      if (n + 1 > size) n = size > 0 ? size - 1 : 0;
      memcpy(arg, c->name, n);
      ((char*)arg)[n] = 0;
      return (int)(n + 1);
    }
    if (nr == 0x07 || nr == 0x08) { /* EVIOCGPHYS / EVIOCGUNIQ */
      if (size) ((char*)arg)[0] = 0;
      return 0;
    }
    if (nr == 0x09 || nr == 0x18 || nr == 0x19 || nr == 0x1a ||
        nr == 0x1b) {            /* EVIOCGPROP / KEY / LED / SND / SW */
      memset(arg, 0, size);
      if (nr == 0x18) {          /* EVIOCGKEY: all buttons up */
      }
      return (int)size;
    }
    if (nr >= 0x20 && nr < 0x40) { /* EVIOCGBIT(ev, len) */
      unsigned evt = nr - 0x20;
      unsigned char* bits = (unsigned char*)arg;
      memset(bits, 0, size);
      if (evt == 0) {            /* supported types: SYN, KEY, ABS */
        if (size > 0) bits[0] |= (1u << 0) | (1u << 1) | (1u << 3);
      } else if (evt == 1) {     /* EV_KEY: bits of btn_map */
        for (unsigned i = 0; i < c->num_btns && i < 64; ++i) {
          unsigned code = c->btn_map[i];
          if (code / 8 < size) bits[code / 8] |= 1u << (code % 8);
        }
      } else if (evt == 3) {     /* EV_ABS: bits of axes_map */
        for (unsigned i = 0; i < c->num_axes && i < 16; ++i) {
          unsigned code = c->axes_map[i];
          if (code / 8 < size) bits[code / 8] |= 1u << (code % 8);
        }
      }
      return (int)size;
    }
    if (nr >= 0x40 && nr < 0x80) { /* EVIOCGABS(abs) */
      int32_t* ai = (int32_t*)arg; /* value,min,max,fuzz,flat,resolution */
      ai[0] = 0;
      ai[1] = -32767;
      ai[2] = 32767;
      ai[3] = 16;
      ai[4] = 128;
      ai[5] = 0;
      return 0;
    }
    if (nr == 0x84) {            /* EVIOCGEFFECTS */
      *(int*)arg = 0;
      return 0;
    }
    if (nr == 0x90)              /* EVIOCGRAB */
      return 0;
    errno = EINVAL;
    return -1;
  }
  if (type != 'j') {
    errno = EINVAL;
    return -1;
  }
  switch (nr) {
    case 0x01: /* JSIOCGVERSION */
      *(uint32_t*)arg = JS_VERSION;
      return 0;
    case 0x11: /* JSIOCGAXES */
      *(uint8_t*)arg = (uint8_t)c->num_axes;
      return 0;
    case 0x12: /* JSIOCGBUTTONS */
      *(uint8_t*)arg = (uint8_t)c->num_btns;
      return 0;
    case 0x13: { /* JSIOCGNAME(len) */
      size_t n = strnlen(c->name, sizeof(c->name));
      if (n + 1 > size) n = size > 0 ? size - 1 : 0;
      memcpy(arg, c->name, n);
      ((char*)arg)[n] = 0;
      return (int)(n + 1);
    }
    case 0x32: { /* JSIOCGAXMAP */
      uint8_t* map = (uint8_t*)arg;
      unsigned n = size < c->num_axes ? size : c->num_axes;
      for (unsigned i = 0; i < n; ++i) map[i] = c->axes_map[i % 16];
      return 0;
    }
    case 0x34: { /* JSIOCGBTNMAP */
      uint16_t* map = (uint16_t*)arg;
      unsigned n = size / 2 < c->num_btns ? size / 2 : c->num_btns;
      for (unsigned i = 0; i < n; ++i) map[i] = c->btn_map[i % 64];
      return 0;
    }
    default:
      errno = EINVAL;
      return -1;
  }
}

int close(int fd) {
  init_real();
  drop_pad(fd);
  return real_close(fd);
}
