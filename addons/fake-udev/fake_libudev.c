/* selkies_amd fake libudev (drop-in libudev.so.1).
 *
 * Fabricates joystick devices for libudev enumeration so containerized
 * apps (SDL etc.) discover the selkies virtual gamepads without kernel
 * devices — the architecture surveyed from the reference fake-udev
 * (SURVEY.md §2.4), re-implemented for OUR interposer: a pad N "exists"
 * while $SELKIES_JS_SOCKET_PATH/selkies_jsN.sock exists; its devnode is
 * /dev/input/jsN (served by the joystick interposer). Hotplug events come
 * from inotify on the socket directory.
 *
 * Implements the libudev calls the common SDL/game enumeration paths use:
 *   udev_new/ref/unref,
 *   udev_enumerate_* (match subsystem, scan, list walk),
 *   udev_device_new_from_syspath + getters (+ parent chain with
 *     ID_INPUT_JOYSTICK), udev_monitor_* (netlink-compatible API backed by
 *     inotify).
 *
 * Build: gcc -O2 -Wall -shared -fPIC -o libudev.so.1 fake_libudev.c
 */
#define _GNU_SOURCE
#include <dirent.h>
#include <limits.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/inotify.h>
#include <unistd.h>

#define MAX_PADS 4

static const char* sock_dir(void) {
  const char* d = getenv("SELKIES_JS_SOCKET_PATH");
  return d ? d : "/tmp/selkies_js";
}

static int pad_present(int idx) {
  char p[PATH_MAX];
  snprintf(p, sizeof(p), "%s/selkies_js%d.sock", sock_dir(), idx);
  return access(p, F_OK) == 0;
}

/* ---------------- object model ---------------- */

struct udev {
  int refs;
};

struct udev_list_entry {
  char name[128];
  char value[128];
  struct udev_list_entry* next;
};

struct udev_device {
  struct udev* udev;
  int refs;
  int pad;              /* 0..3 */
  int is_parent;        /* the "input device" parent w/ properties */
  int is_event;         /* evdev node (event1000+pad) instead of jsN */
  char syspath[192];
  char action[16];
};

struct udev_enumerate {
  struct udev* udev;
  int refs;
  int match_input;
  struct udev_list_entry* list;
};

struct udev_monitor {
  struct udev* udev;
  int refs;
  int ifd;              /* inotify fd */
  int wd;
  int present[MAX_PADS];
};

/* ---------------- core ---------------- */

struct udev* udev_new(void) {
  struct udev* u = calloc(1, sizeof(*u));
  u->refs = 1;
  return u;
}
struct udev* udev_ref(struct udev* u) { if (u) u->refs++; return u; }
struct udev* udev_unref(struct udev* u) {
  if (u && --u->refs == 0) free(u);
  return NULL;
}

/* ---------------- device ---------------- */

static struct udev_device* make_device(struct udev* u, int pad,
                                       int is_parent) {
  struct udev_device* d = calloc(1, sizeof(*d));
  d->udev = u;
  d->refs = 1;
  d->pad = pad;
  d->is_parent = is_parent;
  if (is_parent)
    snprintf(d->syspath, sizeof(d->syspath),
             "/sys/devices/virtual/input/selkies-input%d", pad);
  else
    snprintf(d->syspath, sizeof(d->syspath),
             "/sys/devices/virtual/input/selkies-input%d/js%d", pad, pad);
  return d;
}

static struct udev_device* make_event_device(struct udev* u, int pad) {
  struct udev_device* d = make_device(u, pad, 0);
  d->is_event = 1;
  snprintf(d->syspath, sizeof(d->syspath),
           "/sys/devices/virtual/input/selkies-input%d/event%d", pad,
           1000 + pad);
  return d;
}

struct udev_device* udev_device_new_from_syspath(struct udev* u,
                                                 const char* syspath) {
  int pad;
  if (!syspath) return NULL;
  const char* js = strstr(syspath, "/js");
  if (js && sscanf(js, "/js%d", &pad) == 1 && pad >= 0 && pad < MAX_PADS)
    return make_device(u, pad, 0);
  const char* ev = strstr(syspath, "/event");
  if (ev && sscanf(ev, "/event%d", &pad) == 1 && pad >= 1000 &&
      pad < 1000 + MAX_PADS)
    return make_event_device(u, pad - 1000);
  if (sscanf(syspath, "/sys/devices/virtual/input/selkies-input%d", &pad)
          == 1 && pad >= 0 && pad < MAX_PADS)
    return make_device(u, pad, 1);
  return NULL;
}

struct udev_device* udev_device_ref(struct udev_device* d) {
  if (d) d->refs++;
  return d;
}
struct udev_device* udev_device_unref(struct udev_device* d) {
  if (d && --d->refs == 0) free(d);
  return NULL;
}
struct udev* udev_device_get_udev(struct udev_device* d) { return d->udev; }

const char* udev_device_get_syspath(struct udev_device* d) {
  return d ? d->syspath : NULL;
}
const char* udev_device_get_sysname(struct udev_device* d) {
  if (!d) return NULL;
  static __thread char name[32];
  if (d->is_parent)
    snprintf(name, sizeof(name), "selkies-input%d", d->pad);
  else if (d->is_event)
    snprintf(name, sizeof(name), "event%d", 1000 + d->pad);
  else
    snprintf(name, sizeof(name), "js%d", d->pad);
  return name;
}
const char* udev_device_get_subsystem(struct udev_device* d) {
  (void)d;
  return "input";
}
const char* udev_device_get_devtype(struct udev_device* d) {
  (void)d;
  return NULL;
}
const char* udev_device_get_devnode(struct udev_device* d) {
  if (!d || d->is_parent) return NULL;
  static __thread char node[32];
  if (d->is_event)
    snprintf(node, sizeof(node), "/dev/input/event%d", 1000 + d->pad);
  else
    snprintf(node, sizeof(node), "/dev/input/js%d", d->pad);
  return node;
}
const char* udev_device_get_action(struct udev_device* d) {
  return (d && d->action[0]) ? d->action : NULL;
}

const char* udev_device_get_property_value(struct udev_device* d,
                                           const char* key) {
  if (!d || !key) return NULL;
  if (strcmp(key, "ID_INPUT") == 0 || strcmp(key, "ID_INPUT_JOYSTICK") == 0)
    return "1";
  if (strcmp(key, "ID_BUS") == 0) return "usb";
  if (strcmp(key, "ID_VENDOR_ID") == 0) return "045e";
  if (strcmp(key, "ID_MODEL_ID") == 0) return "028e";
  if (strcmp(key, "DEVNAME") == 0) return udev_device_get_devnode(d);
  if (strcmp(key, "SUBSYSTEM") == 0) return "input";
  return NULL;
}

const char* udev_device_get_sysattr_value(struct udev_device* d,
                                          const char* attr) {
  if (!d || !attr) return NULL;
  if (strcmp(attr, "name") == 0) return "Selkies Virtual Gamepad";
  if (strcmp(attr, "id/vendor") == 0) return "045e";
  if (strcmp(attr, "id/product") == 0) return "028e";
  return NULL;
}

struct udev_device* udev_device_get_parent(struct udev_device* d) {
  if (!d || d->is_parent) return NULL;
  /* note: parent lifetime tied to child per libudev docs; we leak-free via
     child unref not tracking it — acceptable for the shim's use pattern */
  struct udev_device* p = make_device(d->udev, d->pad, 1);
  p->refs = 0; /* owned by child conceptually */
  return p;
}

struct udev_device* udev_device_get_parent_with_subsystem_devtype(
    struct udev_device* d, const char* subsystem, const char* devtype) {
  (void)devtype;
  if (!subsystem || strcmp(subsystem, "input") == 0)
    return udev_device_get_parent(d);
  return NULL;
}

/* ---------------- list walking ---------------- */

struct udev_list_entry* udev_list_entry_get_next(struct udev_list_entry* e) {
  return e ? e->next : NULL;
}
const char* udev_list_entry_get_name(struct udev_list_entry* e) {
  return e ? e->name : NULL;
}
const char* udev_list_entry_get_value(struct udev_list_entry* e) {
  return e ? e->value : NULL;
}

/* ---------------- enumerate ---------------- */

struct udev_enumerate* udev_enumerate_new(struct udev* u) {
  struct udev_enumerate* e = calloc(1, sizeof(*e));
  e->udev = u;
  e->refs = 1;
  return e;
}
struct udev_enumerate* udev_enumerate_ref(struct udev_enumerate* e) {
  if (e) e->refs++;
  return e;
}
static void free_list(struct udev_list_entry* l) {
  while (l) {
    struct udev_list_entry* n = l->next;
    free(l);
    l = n;
  }
}
struct udev_enumerate* udev_enumerate_unref(struct udev_enumerate* e) {
  if (e && --e->refs == 0) {
    free_list(e->list);
    free(e);
  }
  return NULL;
}

int udev_enumerate_add_match_subsystem(struct udev_enumerate* e,
                                       const char* subsystem) {
  if (subsystem && strcmp(subsystem, "input") == 0) e->match_input = 1;
  return 0;
}
int udev_enumerate_add_match_property(struct udev_enumerate* e,
                                      const char* k, const char* v) {
  (void)e; (void)k; (void)v;
  return 0;
}
int udev_enumerate_add_match_sysname(struct udev_enumerate* e,
                                     const char* n) {
  (void)e; (void)n;
  return 0;
}

int udev_enumerate_scan_devices(struct udev_enumerate* e) {
  free_list(e->list);
  e->list = NULL;
  if (!e->match_input) return 0;
  struct udev_list_entry** tail = &e->list;
  for (int i = 0; i < MAX_PADS; ++i) {
    if (!pad_present(i)) continue;
    struct udev_list_entry* ent = calloc(1, sizeof(*ent));
    snprintf(ent->name, sizeof(ent->name),
             "/sys/devices/virtual/input/selkies-input%d/js%d", i, i);
    *tail = ent;
    tail = &ent->next;
    /* the evdev sibling (served by the interposer's event surface) */
    ent = calloc(1, sizeof(*ent));
    snprintf(ent->name, sizeof(ent->name),
             "/sys/devices/virtual/input/selkies-input%d/event%d", i,
             1000 + i);
    *tail = ent;
    tail = &ent->next;
  }
  return 0;
}

struct udev_list_entry* udev_enumerate_get_list_entry(
    struct udev_enumerate* e) {
  return e ? e->list : NULL;
}

/* ---------------- monitor (inotify-backed) ---------------- */

struct udev_monitor* udev_monitor_new_from_netlink(struct udev* u,
                                                   const char* name) {
  (void)name;
  struct udev_monitor* m = calloc(1, sizeof(*m));
  m->udev = u;
  m->refs = 1;
  m->ifd = inotify_init1(IN_NONBLOCK | IN_CLOEXEC);
  return m;
}
struct udev_monitor* udev_monitor_ref(struct udev_monitor* m) {
  if (m) m->refs++;
  return m;
}
struct udev_monitor* udev_monitor_unref(struct udev_monitor* m) {
  if (m && --m->refs == 0) {
    if (m->ifd >= 0) close(m->ifd);
    free(m);
  }
  return NULL;
}
int udev_monitor_filter_add_match_subsystem_devtype(struct udev_monitor* m,
                                                    const char* s,
                                                    const char* d) {
  (void)m; (void)s; (void)d;
  return 0;
}
int udev_monitor_enable_receiving(struct udev_monitor* m) {
  if (m->ifd >= 0 && m->wd == 0)
    m->wd = inotify_add_watch(m->ifd, sock_dir(), IN_CREATE | IN_DELETE);
  for (int i = 0; i < MAX_PADS; ++i) m->present[i] = pad_present(i);
  return 0;
}
int udev_monitor_get_fd(struct udev_monitor* m) { return m->ifd; }
int udev_monitor_set_receive_buffer_size(struct udev_monitor* m, int sz) {
  (void)m; (void)sz;
  return 0;
}

struct udev_device* udev_monitor_receive_device(struct udev_monitor* m) {
  char buf[4096];
  ssize_t n = read(m->ifd, buf, sizeof(buf));
  (void)n;
  /* diff present state to synthesize add/remove actions */
  for (int i = 0; i < MAX_PADS; ++i) {
    int now = pad_present(i);
    if (now != m->present[i]) {
      m->present[i] = now;
      struct udev_device* d = make_device(m->udev, i, 0);
      snprintf(d->action, sizeof(d->action), "%s", now ? "add" : "remove");
      return d;
    }
  }
  return NULL;
}
