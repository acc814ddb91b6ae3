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
static __u64 (*bpf_ktime_get_ns)(void) = (void *)5;
static __u64 (*bpf_get_current_pid_tgid)(void) = (void *)14;
static long (*bpf_get_current_comm)(void *buf, __u32 size) = (void *)16;
static long (*bpf_probe_read_user_str)(void *dst, __u32 size,
                                       const void *unsafe_ptr) = (void *)114;

char LICENSE[] SEC("license") = "GPL";

/* BTF-style ring buffer map (loader: libbpf >= 0.x with BTF support) */
#define __uint(name, val) int(*name)[val]
struct {
  __uint(type, BPF_MAP_TYPE_RINGBUF);
  __uint(max_entries, NERRF_RINGBUF_BYTES);
} events SEC(".maps");

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
  bpf_ringbuf_submit(ev, 0);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_write")
int nerrf_trace_write(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_WRITE);
  if (!ev) return 0;
  /* fd->path resolution needs a kprobe-side fd table map (future work,
   * mirrors the upstream limitation); record the byte count */
  ev->bytes = (__u64)ctx->args[2];
  bpf_ringbuf_submit(ev, 0);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_read")
int nerrf_trace_read(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_READ);
  if (!ev) return 0;
  ev->bytes = (__u64)ctx->args[2];
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
  bpf_ringbuf_submit(ev, 0);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_renameat2")
int nerrf_trace_renameat2(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_RENAME);
  if (!ev) return 0;
  bpf_probe_read_user_str(ev->path, sizeof(ev->path), (void *)ctx->args[1]);
  bpf_probe_read_user_str(ev->new_path, sizeof(ev->new_path),
                          (void *)ctx->args[3]);
  bpf_ringbuf_submit(ev, 0);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_unlinkat")
int nerrf_trace_unlinkat(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_UNLINK);
  if (!ev) return 0;
  bpf_probe_read_user_str(ev->path, sizeof(ev->path), (void *)ctx->args[1]);
  bpf_ringbuf_submit(ev, 0);
  return 0;
}

SEC("tracepoint/syscalls/sys_enter_fchmodat")
int nerrf_trace_fchmodat(struct sys_enter_ctx *ctx) {
  struct nerrf_event *ev = reserve_event(NERRF_SYS_CHMOD);
  if (!ev) return 0;
  bpf_probe_read_user_str(ev->path, sizeof(ev->path), (void *)ctx->args[1]);
  bpf_ringbuf_submit(ev, 0);
  return 0;
}
