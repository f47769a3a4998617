/* volume_cache.so — LD_PRELOAD file-IO interposer.
 *
 * First-party rebuild of the reference's prebuilt volume_cache shim
 * (reference: bin/volume_cache_x86.so injected with VOLUME_CACHE_MAP,
 * pkg/worker/file_cache.go:20-90 — source not in the reference repo):
 * read-path remapping so container reads of volume paths are served from
 * the node-local content cache without the application knowing.
 *
 * VOLUME_CACHE_MAP format: "prefix1=target1;prefix2=target2"
 * A path starting with prefixN is rewritten to targetN + remainder IF the
 * rewritten path exists; otherwise the original path is used (write paths
 * and cache misses fall through untouched).
 */

#define _GNU_SOURCE
#include <dlfcn.h>
#include <errno.h>
#include <fcntl.h>
#include <limits.h>
#include <stdarg.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/stat.h>
#include <unistd.h>

#define MAX_MAPS 16

static struct {
  const char *from;
  size_t from_len;
  const char *to;
} g_maps[MAX_MAPS];
static int g_nmaps = -1; /* -1 = not initialized */

static void vc_init(void) {
  if (g_nmaps >= 0) return;
  g_nmaps = 0;
  const char *spec = getenv("VOLUME_CACHE_MAP");
  if (!spec || !*spec) return;
  char *copy = strdup(spec);
  char *save = NULL;
  for (char *tok = strtok_r(copy, ";", &save); tok && g_nmaps < MAX_MAPS;
       tok = strtok_r(NULL, ";", &save)) {
    char *eq = strchr(tok, '=');
    if (!eq) continue;
    *eq = 0;
    g_maps[g_nmaps].from = strdup(tok);
    g_maps[g_nmaps].from_len = strlen(tok);
    g_maps[g_nmaps].to = strdup(eq + 1);
    g_nmaps++;
  }
  free(copy);
}

/* returns remapped path in buf (and buf) if a mapping applies AND the
 * target exists, else returns the original pointer */
static const char *vc_remap(const char *path, char *buf, size_t buflen) {
  vc_init();
  if (g_nmaps <= 0 || !path || path[0] != '/') return path;
  for (int i = 0; i < g_nmaps; i++) {
    if (strncmp(path, g_maps[i].from, g_maps[i].from_len) == 0 &&
        (path[g_maps[i].from_len] == '/' || path[g_maps[i].from_len] == 0)) {
      snprintf(buf, buflen, "%s%s", g_maps[i].to, path + g_maps[i].from_len);
      struct stat st;
      /* use the raw syscall-ish stat through dlsym'd real stat would
       * recurse; __xstat is messy — plain stat() here resolves to OUR
       * interposer only for the `stat` symbol; use access() via real. */
      typedef int (*access_fn)(const char *, int);
      static access_fn real_access = NULL;
      if (!real_access) real_access = (access_fn)dlsym(RTLD_NEXT, "access");
      (void)st;
      if (real_access && real_access(buf, F_OK) == 0) return buf;
      return path;
    }
  }
  return path;
}

#define REMAP(path) \
  char _vcbuf[PATH_MAX]; \
  path = vc_remap(path, _vcbuf, sizeof _vcbuf);

int open(const char *path, int flags, ...) {
  static int (*real)(const char *, int, ...) = NULL;
  if (!real) real = dlsym(RTLD_NEXT, "open");
  mode_t mode = 0;
  if (flags & O_CREAT) {
    va_list ap;
    va_start(ap, flags);
    mode = va_arg(ap, mode_t);
    va_end(ap);
  }
  if (!(flags & (O_WRONLY | O_RDWR | O_CREAT))) { REMAP(path); }
  return real(path, flags, mode);
}

int open64(const char *path, int flags, ...) {
  static int (*real)(const char *, int, ...) = NULL;
  if (!real) real = dlsym(RTLD_NEXT, "open64");
  mode_t mode = 0;
  if (flags & O_CREAT) {
    va_list ap;
    va_start(ap, flags);
    mode = va_arg(ap, mode_t);
    va_end(ap);
  }
  if (!(flags & (O_WRONLY | O_RDWR | O_CREAT))) { REMAP(path); }
  return real(path, flags, mode);
}

int openat(int dirfd, const char *path, int flags, ...) {
  static int (*real)(int, const char *, int, ...) = NULL;
  if (!real) real = dlsym(RTLD_NEXT, "openat");
  mode_t mode = 0;
  if (flags & O_CREAT) {
    va_list ap;
    va_start(ap, flags);
    mode = va_arg(ap, mode_t);
    va_end(ap);
  }
  if (path && path[0] == '/' && !(flags & (O_WRONLY | O_RDWR | O_CREAT))) {
    REMAP(path);
  }
  return real(dirfd, path, flags, mode);
}

FILE *fopen(const char *path, const char *mode) {
  static FILE *(*real)(const char *, const char *) = NULL;
  if (!real) real = dlsym(RTLD_NEXT, "fopen");
  if (mode && mode[0] == 'r') { REMAP(path); }
  return real(path, mode);
}

FILE *fopen64(const char *path, const char *mode) {
  static FILE *(*real)(const char *, const char *) = NULL;
  if (!real) real = dlsym(RTLD_NEXT, "fopen64");
  if (mode && mode[0] == 'r') { REMAP(path); }
  return real(path, mode);
}

int stat(const char *path, struct stat *st) {
  static int (*real)(const char *, struct stat *) = NULL;
  if (!real) real = dlsym(RTLD_NEXT, "stat");
  REMAP(path);
  return real(path, st);
}

int lstat(const char *path, struct stat *st) {
  static int (*real)(const char *, struct stat *) = NULL;
  if (!real) real = dlsym(RTLD_NEXT, "lstat");
  REMAP(path);
  return real(path, st);
}

int access(const char *path, int mode) {
  static int (*real)(const char *, int) = NULL;
  if (!real) real = dlsym(RTLD_NEXT, "access");
  REMAP(path);
  return real(path, mode);
}

/* glibc's stat on some paths goes through __xstat */
int __xstat(int ver, const char *path, struct stat *st) {
  static int (*real)(int, const char *, struct stat *) = NULL;
  if (!real) real = dlsym(RTLD_NEXT, "__xstat");
  if (!real) { errno = ENOSYS; return -1; }
  REMAP(path);
  return real(ver, path, st);
}
