// Declarations for X11 extension client APIs whose dev headers are absent
// from this image (libXext/libXtst runtime .so's ARE present). These
// prototypes/struct layouts are the long-stable public ABI of libXext
// (MIT-SHM, X11R6+) and libXtst; declaring them here avoids a build
// dependency without changing behavior.
#pragma once

#include <X11/Xlib.h>

extern "C" {

// ---- MIT-SHM (libXext) ----
typedef unsigned long ShmSeg;

typedef struct {
  ShmSeg shmseg;
  int shmid;
  char* shmaddr;
  Bool readOnly;
} XShmSegmentInfo;

Bool XShmQueryExtension(Display*);
XImage* XShmCreateImage(Display*, Visual*, unsigned int depth, int format,
                        char* data, XShmSegmentInfo*, unsigned int width,
                        unsigned int height);
Bool XShmAttach(Display*, XShmSegmentInfo*);
Bool XShmDetach(Display*, XShmSegmentInfo*);
Bool XShmGetImage(Display*, Drawable, XImage*, int x, int y,
                  unsigned long plane_mask);

// ---- XTEST (libXtst) ----
Bool XTestQueryExtension(Display*, int* event_base, int* error_base,
                         int* major, int* minor);
int XTestFakeKeyEvent(Display*, unsigned int keycode, Bool is_press,
                      unsigned long delay);
int XTestFakeButtonEvent(Display*, unsigned int button, Bool is_press,
                         unsigned long delay);
int XTestFakeMotionEvent(Display*, int screen, int x, int y,
                         unsigned long delay);
int XTestFakeRelativeMotionEvent(Display*, int x, int y, unsigned long delay);

// ---- XFIXES (libXfixes) cursor fetch ----
typedef struct {
  short x, y;
  unsigned short width, height;
  unsigned short xhot, yhot;
  unsigned long cursor_serial;
  unsigned long* pixels;  // ARGB, unsigned long per pixel
  Atom atom;
  const char* name;
} XFixesCursorImage;

Bool XFixesQueryExtension(Display*, int* event_base, int* error_base);
XFixesCursorImage* XFixesGetCursorImage(Display*);

}  // extern "C"
