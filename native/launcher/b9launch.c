/* b9launch — minimal first-party container launcher for MI355X nodes.
 *
 * The runc-equivalent of this framework (reference: beta9's patched runc,
 * pkg/runtime/runc.go; built from source here because the target image has
 * no runc). Provides:
 *   - mount + PID + UTS + IPC namespaces (CLONE_NEWNS|NEWPID|NEWUTS|NEWIPC)
 *   - optional network namespace for --block-network (no veth: the
 *     container then simply has no network, which is the block semantics)
 *   - overlayfs root (lowerdir = host rootfs or an image tree, tmpfs-backed
 *     upper/work), pivot_root, fresh /proc, minimal /dev with ONLY the
 *     device nodes passed via --dev (this is the AMD GPU isolation story:
 *     /dev/kfd + the assigned /dev/dri/renderD* and nothing else)
 *   - cgroup v2 cpu.max / memory.max limits
 *   - signal forwarding and exit-code propagation
 *
 * Usage:
 *   b9launch [--rootfs-overlay DIR] [--cpu MILLICORES] [--mem MB]
 *            [--hostname NAME] [--dev /dev/node]... [--bind SRC:DST]...
 *            [--env K=V]... [--cwd DIR] [--block-network]
 *            [--lower DIR1:DIR2:...] [--upper DIR --work DIR]
 *            [--cgroup-parent NAME] -- CMD [ARGS...]
 *
 * --rootfs-overlay DIR: DIR must contain (or will get) upper/ work/ merged/;
 * lowerdir is "/" unless DIR/lower exists (a prepared image tree).
 * --lower: explicit overlayfs lowerdir chain (topmost first) — image layers
 *   from the content-addressed layer store.
 * --upper/--work: host-disk-backed upper/work instead of the tmpfs scratch,
 *   so a build container's filesystem delta SURVIVES exit and can be
 *   captured as a new image layer (reference: build containers +
 *   captureCheckpointFilesystem upper-layer semantics).
 */

#define _GNU_SOURCE
#include <errno.h>
#include <fcntl.h>
#include <sched.h>
#include <signal.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/mount.h>
#include <sys/stat.h>
#include <sys/syscall.h>
#include <sys/types.h>
#include <sys/prctl.h>
#include <sys/wait.h>
#include <unistd.h>

#define MAX_DEVS 32
#define MAX_BINDS 32
#define MAX_ENVS 128

static const char *g_overlay_dir = NULL;
static long g_cpu_milli = 0;
static long g_mem_mb = 0;
static const char *g_hostname = NULL;
static const char *g_devs[MAX_DEVS];
static int g_ndevs = 0;
static const char *g_binds[MAX_BINDS];
static int g_nbinds = 0;
static const char *g_masks[MAX_BINDS];  /* paths hidden behind empty tmpfs */
static int g_nmasks = 0;
static char *g_envs[MAX_ENVS];
static int g_nenvs = 0;
static const char *g_cwd = NULL;
static int g_block_network = 0;
static const char *g_cgroup_parent = "b9";
static const char *g_lower = NULL;   /* explicit lowerdir chain (topmost first) */
static const char *g_upper = NULL;   /* host-backed upper (build capture) */
static const char *g_work = NULL;    /* host-backed workdir */
static int g_netns_fd = -1;          /* join a preallocated network namespace */
static char **g_cmd = NULL;

static void die(const char *msg) {
  fprintf(stderr, "b9launch: %s: %s\n", msg, strerror(errno));
  exit(125);
}

static void write_file(const char *path, const char *data) {
  int fd = open(path, O_WRONLY);
  if (fd < 0) return; /* best-effort (controller may be absent) */
  ssize_t n = write(fd, data, strlen(data));
  (void)n;
  close(fd);
}

static int setup_cgroup(pid_t pid) {
  if (g_cpu_milli <= 0 && g_mem_mb <= 0) return 0;
  char dir[512], path[640], buf[128];
  snprintf(dir, sizeof dir, "/sys/fs/cgroup/%s", g_cgroup_parent);
  mkdir(dir, 0755);
  snprintf(dir, sizeof dir, "/sys/fs/cgroup/%s/c%d", g_cgroup_parent, pid);
  if (mkdir(dir, 0755) != 0 && errno != EEXIST) return -1;
  if (g_mem_mb > 0) {
    snprintf(path, sizeof path, "%s/memory.max", dir);
    snprintf(buf, sizeof buf, "%ld", g_mem_mb * 1024 * 1024);
    write_file(path, buf);
  }
  if (g_cpu_milli > 0) {
    snprintf(path, sizeof path, "%s/cpu.max", dir);
    snprintf(buf, sizeof buf, "%ld 100000", g_cpu_milli * 100);
    write_file(path, buf);
  }
  snprintf(path, sizeof path, "%s/cgroup.procs", dir);
  snprintf(buf, sizeof buf, "%d", pid);
  write_file(path, buf);
  return 0;
}

static void cleanup_cgroup(pid_t pid) {
  char dir[512];
  snprintf(dir, sizeof dir, "/sys/fs/cgroup/%s/c%d", g_cgroup_parent, pid);
  rmdir(dir);
}

static void bind_dev(const char *merged, const char *node) {
  /* create the node path under merged/dev and bind-mount the host node */
  char dst[1024];
  if (strncmp(node, "/dev/", 5) != 0) return;
  snprintf(dst, sizeof dst, "%s%s", merged, node);
  /* mkdir -p parents */
  char tmp[1024];
  snprintf(tmp, sizeof tmp, "%s", dst);
  for (char *p = tmp + strlen(merged) + 1; *p; p++) {
    if (*p == '/') {
      *p = 0;
      mkdir(tmp, 0755);
      *p = '/';
    }
  }
  int fd = open(dst, O_CREAT | O_WRONLY, 0666);
  if (fd >= 0) close(fd);
  if (mount(node, dst, NULL, MS_BIND, NULL) != 0)
    fprintf(stderr, "b9launch: bind %s failed: %s\n", node, strerror(errno));
}

static pid_t g_payload = -1;
static void forward_to_payload(int sig) {
  if (g_payload > 0) kill(g_payload, sig);
}

static int child_main(void *arg) {
  (void)arg;
  /* if the host-side wrapper dies (e.g. SIGKILL), take the whole
   * namespace down with it — otherwise payloads leak unkillable */
  prctl(PR_SET_PDEATHSIG, SIGKILL);
  if (getppid() == 1) _exit(1); /* wrapper already gone before prctl */
  if (g_netns_fd >= 0) {
    /* join the preallocated slot namespace (veth/IP already programmed by
     * the worker's network plane) — the slot-pool fast path */
    if (setns(g_netns_fd, CLONE_NEWNET) != 0)
      die("setns netns");
    close(g_netns_fd);
  }
  if (g_hostname && sethostname(g_hostname, strlen(g_hostname)) != 0)
    fprintf(stderr, "b9launch: sethostname: %s\n", strerror(errno));

  /* never propagate our mounts back to the host */
  if (mount(NULL, "/", NULL, MS_REC | MS_PRIVATE, NULL) != 0)
    die("make / rprivate");

  if (g_overlay_dir) {
    char upper[768], work[768], merged[768], lower[768], opts[4000];
    mkdir(g_overlay_dir, 0755);
    if (g_upper && g_work) {
      /* host-disk-backed upper/work (build capture): no tmpfs scratch */
      snprintf(upper, sizeof upper, "%s", g_upper);
      snprintf(work, sizeof work, "%s", g_work);
      snprintf(merged, sizeof merged, "%s/merged", g_overlay_dir);
      mkdir(upper, 0755);
      mkdir(work, 0755);
      mkdir(merged, 0755);
    } else {
      snprintf(upper, sizeof upper, "%s/upper", g_overlay_dir);
      snprintf(work, sizeof work, "%s/work", g_overlay_dir);
      snprintf(merged, sizeof merged, "%s/merged", g_overlay_dir);
      /* upper/work must be on tmpfs so nesting on any host fs works */
      if (mount("tmpfs", g_overlay_dir, "tmpfs", 0, "size=90%") != 0)
        die("tmpfs for overlay scratch");
      mkdir(upper, 0755);
      mkdir(work, 0755);
      mkdir(merged, 0755);
    }
    snprintf(lower, sizeof lower, "%s/lower", g_overlay_dir);
    struct stat st;
    const char *lowerdir;
    if (g_lower) lowerdir = g_lower;            /* image layer chain */
    else lowerdir = (stat(lower, &st) == 0) ? lower : "/";
    snprintf(opts, sizeof opts, "lowerdir=%s,upperdir=%s,workdir=%s",
             lowerdir, upper, work);
    if (mount("overlay", merged, "overlay", 0, opts) != 0) die("overlay mount");

    /* fresh /proc + sysfs (ROCm user space needs /sys/class/kfd and the
     * amdgpu sysfs tree), minimal /dev with only the allowed nodes */
    char path[1024];
    snprintf(path, sizeof path, "%s/proc", merged);
    mkdir(path, 0555);
    if (mount("proc", path, "proc", 0, NULL) != 0) die("proc mount");
    snprintf(path, sizeof path, "%s/sys", merged);
    mkdir(path, 0555);
    if (mount("sysfs", path, "sysfs", 0, NULL) != 0) {
      /* fall back to a read-only bind of the host /sys (needed when a
       * network namespace denies a fresh sysfs mount) */
      if (mount("/sys", path, NULL, MS_BIND | MS_REC, NULL) == 0)
        mount(NULL, path, NULL, MS_REMOUNT | MS_BIND | MS_RDONLY | MS_REC, NULL);
    }
    snprintf(path, sizeof path, "%s/dev", merged);
    mkdir(path, 0755);
    if (mount("tmpfs", path, "tmpfs", 0, "mode=755,size=16m") != 0)
      die("dev tmpfs");
    const char *std_devs[] = {"/dev/null", "/dev/zero", "/dev/full",
                              "/dev/random", "/dev/urandom", "/dev/tty"};
    for (unsigned i = 0; i < sizeof std_devs / sizeof *std_devs; i++)
      bind_dev(merged, std_devs[i]);
    for (int i = 0; i < g_ndevs; i++) bind_dev(merged, g_devs[i]);
    snprintf(path, sizeof path, "%s/dev/shm", merged);
    mkdir(path, 0777);
    if (mount("tmpfs", path, "tmpfs", 0, "mode=1777") != 0) die("shm mount");
    snprintf(path, sizeof path, "%s/dev/pts", merged);
    mkdir(path, 0755);
    mount("devpts", path, "devpts", 0, "newinstance,ptmxmode=0666");
    /* ptys (sandbox terminals) open /dev/ptmx */
    snprintf(path, sizeof path, "%s/dev/ptmx", merged);
    if (symlink("pts/ptmx", path) != 0 && errno != EEXIST)
      fprintf(stderr, "b9launch: ptmx symlink: %s\n", strerror(errno));
    snprintf(path, sizeof path, "%s/tmp", merged);
    mkdir(path, 01777);
    mount("tmpfs", path, "tmpfs", 0, "mode=1777");

    /* masks: hide host data roots (gateway DB, checkpoints, workspaces)
     * behind empty read-only tmpfs — the host-rootfs lowerdir must not
     * leak other tenants' state. Binds run AFTER masks, so a specific
     * volume/image dir can be re-exposed inside a masked root. */
    for (int i = 0; i < g_nmasks; i++) {
      struct stat mst;
      snprintf(path, sizeof path, "%s%s", merged, g_masks[i]);
      if (stat(path, &mst) != 0 || !S_ISDIR(mst.st_mode)) continue;
      /* small+writable so later --bind targets can still be created under
       * a masked root (e.g. the image store inside the data root) */
      if (mount("tmpfs", path, "tmpfs", 0, "mode=755,size=1m") != 0)
        fprintf(stderr, "b9launch: mask %s failed: %s\n", g_masks[i],
                strerror(errno));
    }

    /* user-requested bind mounts (volumes, workspace dirs): SRC:DST */
    for (int i = 0; i < g_nbinds; i++) {
      char spec[1024];
      snprintf(spec, sizeof spec, "%s", g_binds[i]);
      char *colon = strchr(spec, ':');
      if (!colon) continue;
      *colon = 0;
      const char *src = spec, *dstrel = colon + 1;
      snprintf(path, sizeof path, "%s%s", merged, dstrel);
      char mk[1024];
      snprintf(mk, sizeof mk, "%s", path);
      for (char *p = mk + strlen(merged) + 1; *p; p++) {
        if (*p == '/') { *p = 0; mkdir(mk, 0755); *p = '/'; }
      }
      mkdir(path, 0755);
      if (mount(src, path, NULL, MS_BIND | MS_REC, NULL) != 0)
        fprintf(stderr, "b9launch: bind %s -> %s failed: %s\n", src, dstrel,
                strerror(errno));
    }

    /* pivot into the merged rootfs */
    snprintf(path, sizeof path, "%s/.oldroot", merged);
    mkdir(path, 0700);
    if (syscall(SYS_pivot_root, merged, path) != 0) die("pivot_root");
    if (chdir("/") != 0) die("chdir /");
    if (umount2("/.oldroot", MNT_DETACH) != 0) die("umount oldroot");
    rmdir("/.oldroot");
  } else {
    /* shared-rootfs mode: still isolate /proc for the new PID ns */
    if (mount("proc", "/proc", "proc", 0, NULL) != 0)
      fprintf(stderr, "b9launch: proc remount: %s\n", strerror(errno));
  }

  for (int i = 0; i < g_nenvs; i++) putenv(g_envs[i]);
  if (g_cwd && chdir(g_cwd) != 0)
    fprintf(stderr, "b9launch: chdir %s: %s\n", g_cwd, strerror(errno));

  /* mini-init: we are PID 1 in the namespace, where default-disposition
   * signals are IGNORED — exec'ing the payload directly would make every
   * SIGTERM a no-op (stops would always escalate to SIGKILL). Fork the
   * payload, forward signals, reap orphans (tini behavior). */
  g_payload = fork();
  if (g_payload < 0) die("fork payload");
  if (g_payload == 0) {
    execvp(g_cmd[0], g_cmd);
    fprintf(stderr, "b9launch: exec %s: %s\n", g_cmd[0], strerror(errno));
    _exit(127);
  }
  signal(SIGTERM, forward_to_payload);
  signal(SIGINT, forward_to_payload);
  signal(SIGQUIT, forward_to_payload);
  signal(SIGHUP, forward_to_payload);
  for (;;) {
    int st;
    pid_t p = wait(&st);
    if (p == g_payload) {
      if (WIFEXITED(st)) return WEXITSTATUS(st);
      if (WIFSIGNALED(st)) return 128 + WTERMSIG(st);
      return 1;
    }
    if (p < 0) {
      if (errno == EINTR) continue;
      return 1; /* no children left (shouldn't happen before payload exit) */
    }
    /* reaped an orphan; keep waiting */
  }
}

static pid_t g_child = -1;
static void forward_signal(int sig) {
  if (g_child > 0) kill(g_child, sig);
}

int main(int argc, char **argv) {
  int i = 1;
  for (; i < argc; i++) {
    if (strcmp(argv[i], "--") == 0) { i++; break; }
    else if (strcmp(argv[i], "--rootfs-overlay") == 0) g_overlay_dir = argv[++i];
    else if (strcmp(argv[i], "--cpu") == 0) g_cpu_milli = atol(argv[++i]);
    else if (strcmp(argv[i], "--mem") == 0) g_mem_mb = atol(argv[++i]);
    else if (strcmp(argv[i], "--hostname") == 0) g_hostname = argv[++i];
    else if (strcmp(argv[i], "--dev") == 0 && g_ndevs < MAX_DEVS) g_devs[g_ndevs++] = argv[++i];
    else if (strcmp(argv[i], "--bind") == 0 && g_nbinds < MAX_BINDS) g_binds[g_nbinds++] = argv[++i];
    else if (strcmp(argv[i], "--mask") == 0 && g_nmasks < MAX_BINDS) g_masks[g_nmasks++] = argv[++i];
    else if (strcmp(argv[i], "--env") == 0 && g_nenvs < MAX_ENVS) g_envs[g_nenvs++] = argv[++i];
    else if (strcmp(argv[i], "--cwd") == 0) g_cwd = argv[++i];
    else if (strcmp(argv[i], "--block-network") == 0) g_block_network = 1;
    else if (strcmp(argv[i], "--netns-fd") == 0) g_netns_fd = atoi(argv[++i]);
    else if (strcmp(argv[i], "--lower") == 0) g_lower = argv[++i];
    else if (strcmp(argv[i], "--upper") == 0) g_upper = argv[++i];
    else if (strcmp(argv[i], "--work") == 0) g_work = argv[++i];
    else if (strcmp(argv[i], "--cgroup-parent") == 0) g_cgroup_parent = argv[++i];
    else { fprintf(stderr, "b9launch: unknown arg %s\n", argv[i]); return 125; }
  }
  if (i >= argc) { fprintf(stderr, "b9launch: no command\n"); return 125; }
  g_cmd = &argv[i];

  int flags = CLONE_NEWNS | CLONE_NEWPID | CLONE_NEWUTS | CLONE_NEWIPC | SIGCHLD;
  if (g_block_network) flags |= CLONE_NEWNET;

  const long STACK = 1024 * 1024;
  char *stack = malloc(STACK);
  if (!stack) die("malloc stack");
  g_child = clone(child_main, stack + STACK, flags, NULL);
  if (g_child < 0) die("clone");

  setup_cgroup(g_child);
  signal(SIGTERM, forward_signal);
  signal(SIGINT, forward_signal);
  signal(SIGQUIT, forward_signal);

  int status = 0;
  while (waitpid(g_child, &status, 0) < 0 && errno == EINTR) {}
  cleanup_cgroup(g_child);
  if (WIFEXITED(status)) return WEXITSTATUS(status);
  if (WIFSIGNALED(status)) return 128 + WTERMSIG(status);
  return 1;
}
