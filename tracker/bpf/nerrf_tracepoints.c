/* nerrf-amd eBPF syscall capture — tracepoint programs.
 *
 * Hooks sys_enter_{openat,write,read,rename,renameat2,unlinkat,fchmodat}
 * and pushes fixed-size nerrf_event records through a BPF ring buffer with
 * drop-on-full semantics (capture must never block the traced workload —
 * same operational contract as the upstream tracker, extended with the
 * read/unlink/chmod hooks its docs plan as M2).
 *
 * Built header-free (`clang -O2 -target bpf`): this image has no libbpf
 * headers, so the helper stubs and tracepoint context layout are declared
 * locally.  Requires kernel >= 5.8 (BPF_MAP_TYPE_RINGBUF).
 */
#include <linux/bpf.h>
#include <linux/types.h>

#include "event_abi.h"

#define SEC(name) __attribute__((section(name), used))

/* helper stubs by ABI id (bpf_helper_defs.h is not shipped here) */
static void *(*bpf_ringbuf_reserve)(void *ringbuf, __u64 size, __u64 flags) =
    (void *)131;
static void (*bpf_ringbuf_submit)(void *data, __u64 flags) = (void *)132;
static void (*bpf_ringbuf_discard)(void *data, __u64 flags) = (void *)133;
static __u64 (*bpf_ktime_get_ns)(void) = (void *)5;
static __u64 (*bpf_get_current_pid_tgid)(void) = (void *)14;
static long (*bpf_get_current_comm)(void *buf, __u32 size) = (void *)16;
static long (*bpf_probe_read_user_str)(void *dst, __u32 size,
                                       const void *unsafe_ptr) = (void *)114;
static void *(*bpf_map_lookup_elem)(void *map, const void *key) = (void *)1;
static long (*bpf_map_update_elem)(void *map, const void *key,
                                   const void *value, __u64 flags) = (void *)2;
static long (*bpf_map_delete_elem)(void *map, const void *key) = (void *)3;

char LICENSE[] SEC("license") = "GPL";

/* BTF-style ring buffer map (loader: libbpf >= 0.x with BTF support) */
#define __uint(name, val) int(*name)[val]
#define __type(name, val) typeof(val) *name
struct {
  __uint(type, BPF_MAP_TYPE_RINGBUF);
  __uint(max_entries, NERRF_RINGBUF_BYTES);
} events SEC(".maps");

/* fd -> path resolution for write/read (reference M2 spec,
 * implementation.mdx:520-563: "per-thread fd->path map, registered at
 * openat time").  Two hash maps:
 *   pending_open : tid -> path, written at sys_enter_openat, bound to the
 *                  returned fd at sys_exit_openat;
 *   fd_paths     : (tgid << 32 | fd) -> path, consumed by write/read and
 *                  dropped on close.
 * Bounded (LRU semantics via plain hash + delete-on-close; stale entries
 * for long-lived fds are the accepted trade — same as the upstream plan).
 */
struct fd_path_val {
  char path[NERRF_PATH_MAX];
};

struct {
  __uint(type, BPF_MAP_TYPE_HASH);
  __uint(max_entries, 8192);
  __type(key, __u32);
  __type(value, struct fd_path_val);
} pending_open SEC(".maps");

struct {
  __uint(type, BPF_MAP_TYPE_HASH);
  __uint(max_entries, 65536);
  __type(key, __u64);
  __type(value, struct fd_path_val);
} fd_paths SEC(".maps");

struct sys_exit_ctx {
  __u64 _pad;
  long id;
  long ret;
};

/* BPF-side path filtering (upstream M2 plan, overview.mdx:262-269): when
 * the daemon programs prefixes into this map, events whose paths match
 * none of them are DISCARDED in-kernel — the ring buffer and userspace
 * never see off-scope traffic.  Pathless events (plain write/read before
 * fd resolution) always pass. */
#define NERRF_FILTER_MAX 8
#define NERRF_FILTER_LEN 64
struct nerrf_filter_cfg {
  __u32 n;
  char prefix[NERRF_FILTER_MAX][NERRF_FILTER_LEN];
};

struct {
  __uint(type, BPF_MAP_TYPE_ARRAY);
  __uint(max_entries, 1);
  __type(key, __u32);
  __type(value, struct nerrf_filter_cfg);
} path_filter SEC(".maps");

static __always_inline int nerrf_path_allowed(const char *path) {
  __u32 zero = 0;
  struct nerrf_filter_cfg *cfg = bpf_map_lookup_elem(&path_filter, &zero);
  if (!cfg || cfg->n == 0) return 1;
  if (path[0] == 0) return 1;
#pragma unroll
  for (int i = 0; i < NERRF_FILTER_MAX; i++) {
    if (i >= (int)cfg->n) break;
    int match = 1;
#pragma unroll
    for (int j = 0; j < NERRF_FILTER_LEN; j++) {
      char c = cfg->prefix[i][j];
      if (c == 0) break;
      if (path[j] != c) { match = 0; break; }
    }
    if (match) return 1;
  }
  return 0;
}

/* submit unless the filter rejects the event's path(s) */
static __always_inline void submit_filtered(struct nerrf_event *ev) {
  if (nerrf_path_allowed(ev->path) ||
      (ev->new_path[0] && nerrf_path_allowed(ev->new_path)))
    bpf_ringbuf_submit(ev, 0);
  else
    bpf_ringbuf_discard(ev, 0);
}

/* raw tracepoint context for syscalls:sys_enter_* */
struct sys_enter_ctx {
  __u64 _pad;
  long id;
  unsigned long args[6];
};

static __always_inline struct nerrf_event *reserve_event(int syscall_id) {
  struct nerrf_event *ev =
      bpf_ringbuf_reserve(&events, sizeof(struct nerrf_event), 0);
  if (!ev) return 0; /* ring full: drop silently, never block the workload */
  __u64 id = bpf_get_current_pid_tgid();
  ev->ts_ns = bpf_ktime_get_ns();
  ev->pid = id >> 32;
  ev->tid = (__u32)id;
  ev->syscall_id = syscall_id;
  ev->flags = 0;
  ev->ret_val = 0;
  ev->bytes = 0;
  ev->path[0] = 0;
  ev->new_path[0] = 0;
  bpf_get_current_comm(ev->comm, sizeof(ev->comm));
  return ev;
}

SEC("tracepoint/syscalls/sys_enter_openat")
int nerrf_trace_openat(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_OPENAT);
  if (!ev) return 0;
  bpf_probe_read_user_str(ev->path, sizeof(ev->path), (void *)ctx->args[1]);
  ev->flags = (int)(ctx->args[2] & 3); /* O_ACCMODE */
  /* stage the path for fd binding at sys_exit_openat */
  {
    __u32 tid = (__u32)bpf_get_current_pid_tgid();
    struct fd_path_val v = {};
    bpf_probe_read_user_str(v.path, sizeof(v.path), (void *)ctx->args[1]);
    bpf_map_update_elem(&pending_open, &tid, &v, 0 /* BPF_ANY */);
  }
  submit_filtered(ev);
  return 0;
}

SEC("tracepoint/syscalls/sys_exit_openat")
int nerrf_trace_openat_exit(struct sys_exit_ctx *ctx) {
  __u64 id = bpf_get_current_pid_tgid();
  __u32 tid = (__u32)id;
  struct fd_path_val *v = bpf_map_lookup_elem(&pending_open, &tid);
  if (!v) return 0;
  if (ctx->ret >= 0) {
    __u64 key = ((id >> 32) << 32) | (__u64)(__u32)ctx->ret;
    bpf_map_update_elem(&fd_paths, &key, v, 0);
  }
  bpf_map_delete_elem(&pending_open, &tid);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_close")
int nerrf_trace_close(struct sys_enter_ctx *ctx) {
  __u64 id = bpf_get_current_pid_tgid();
  __u64 key = ((id >> 32) << 32) | (__u64)(__u32)ctx->args[0];
  bpf_map_delete_elem(&fd_paths, &key);
  return 0;
}

static __always_inline void fill_fd_path(struct nerrf_event *ev, __u64 fd) {
  __u64 id = bpf_get_current_pid_tgid();
  __u64 key = ((id >> 32) << 32) | (fd & 0xFFFFFFFF);
  struct fd_path_val *v = bpf_map_lookup_elem(&fd_paths, &key);
  if (v) __builtin_memcpy(ev->path, v->path, NERRF_PATH_MAX);
}

SEC("tracepoint/syscalls/sys_enter_write")
int nerrf_trace_write(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_WRITE);
  if (!ev) return 0;
  ev->bytes = (__u64)ctx->args[2];
  fill_fd_path(ev, ctx->args[0]); /* resolved via the openat-time fd map */
  bpf_ringbuf_submit(ev, 0);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_read")
int nerrf_trace_read(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_READ);
  if (!ev) return 0;
  ev->bytes = (__u64)ctx->args[2];
  fill_fd_path(ev, ctx->args[0]);
  bpf_ringbuf_submit(ev, 0);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_rename")
int nerrf_trace_rename(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_RENAME);
  if (!ev) return 0;
  bpf_probe_read_user_str(ev->path, sizeof(ev->path), (void *)ctx->args[0]);
  bpf_probe_read_user_str(ev->new_path, sizeof(ev->new_path),
                          (void *)ctx->args[1]);
  submit_filtered(ev);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_renameat2")
int nerrf_trace_renameat2(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_RENAME);
  if (!ev) return 0;
  bpf_probe_read_user_str(ev->path, sizeof(ev->path), (void *)ctx->args[1]);
  bpf_probe_read_user_str(ev->new_path, sizeof(ev->new_path),
                          (void *)ctx->args[3]);
  submit_filtered(ev);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_unlinkat")
int nerrf_trace_unlinkat(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_UNLINK);
  if (!ev) return 0;
  bpf_probe_read_user_str(ev->path, sizeof(ev->path), (void *)ctx->args[1]);
  submit_filtered(ev);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_fchmodat")
int nerrf_trace_fchmodat(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_CHMOD);
  if (!ev) return 0;
  bpf_probe_read_user_str(ev->path, sizeof(ev->path), (void *)ctx->args[1]);
  submit_filtered(ev);
  return 0;
}
