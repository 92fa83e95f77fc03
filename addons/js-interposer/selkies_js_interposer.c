/* selkies_amd joystick interposer (LD_PRELOAD).
 *
 * Redirects opens of /dev/input/jsN to the selkies gamepad unix sockets
 * (SELKIES_JS_SOCKET_PATH/selkies_jsN.sock) so containerized apps see
 * gamepads without kernel devices — the architecture surveyed from the
 * reference js-interposer, re-implemented against OUR
 * socket protocol (selkies_amd/gamepad.py):
 *   on connect the server sends a JsConfig struct:
 *     magic "SJSG", u16 version, u16 vendor, u16 product, u16 num_btns,
 *     u16 num_axes, char name[128], u16 btn_map[64], u8 axes_map[16]
 *   followed by a stream of `struct js_event` records (joydev ABI), so
 *   read() passes straight through.
 *
 * Interposes: open, open64, openat, ioctl, close, access.
 * Emulated joydev ioctls: JSIOCGVERSION, JSIOCGAXES, JSIOCGBUTTONS,
 * JSIOCGNAME, JSIOCGAXMAP, JSIOCGBTNMAP.
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
  int fd;                 /* socket fd doubling as the joydev fd */
  js_config_t config;
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

static void init_real(void) {
  if (!real_open) real_open = dlsym(RTLD_NEXT, "open");
  if (!real_open64) real_open64 = dlsym(RTLD_NEXT, "open64");
  if (!real_openat) real_openat = dlsym(RTLD_NEXT, "openat");
  if (!real_ioctl) real_ioctl = dlsym(RTLD_NEXT, "ioctl");
  if (!real_close) real_close = dlsym(RTLD_NEXT, "close");
  if (!real_access) real_access = dlsym(RTLD_NEXT, "access");
}

/* returns pad index for /dev/input/jsN (N < MAX_PADS), else -1 */
static int js_index(const char* path) {
  int n;
  if (!path) return -1;
  if (sscanf(path, "/dev/input/js%d", &n) == 1 && n >= 0 && n < MAX_PADS)
    return n;
  return -1;
}

static ssize_t read_full(int fd, void* buf, size_t n) {
  size_t got = 0;
  while (got < n) {
    ssize_t r = read(fd, (char*)buf + got, n - got);
    if (r <= 0) {
      if (r < 0 && (errno == EINTR)) continue;
      return -1;
    }
    got += (size_t)r;
  }
  return (ssize_t)got;
}

static int open_pad(int idx) {
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
    g_pads[g_npads].fd = fd;
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
  if (idx >= 0) return open_pad(idx);
  va_list ap;
  va_start(ap, flags);
  mode_t mode = va_arg(ap, mode_t);
  va_end(ap);
  return real_open(path, flags, mode);
}

int open64(const char* path, int flags, ...) {
  init_real();
  int idx = js_index(path);
  if (idx >= 0) return open_pad(idx);
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
  if (idx >= 0) return open_pad(idx);
  va_list ap;
  va_start(ap, flags);
  mode_t mode = va_arg(ap, mode_t);
  va_end(ap);
  return real_openat(dirfd, path, flags, mode);
}

int access(const char* path, int mode) {
  init_real();
  if (js_index(path) >= 0) return 0; /* pretend the device exists */
  return real_access(path, mode);
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
