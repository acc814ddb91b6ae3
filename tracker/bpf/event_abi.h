/* Kernel <-> userspace event ABI for the nerrf-amd tracker.
 *
 * Fixed-size little-endian record pushed through the BPF ring buffer; the
 * collector daemon (tracker/daemon/nerrfd.cpp) parses exactly this layout.
 * Grown relative to the upstream contract (3 syscalls, 564 B) to cover the
 * documented M2 extensions (unlink/chmod hooks — reference docs
 * tracker/implementation.mdx:520-563) while keeping the same
 * drop-on-full ring-buffer semantics.
 */
#ifndef NERRF_EVENT_ABI_H
#define NERRF_EVENT_ABI_H

#define NERRF_PATH_MAX 256
#define NERRF_COMM_LEN 16

/* syscall ids — must match nerrf_amd/data/trace.py SYSCALL_IDS */
enum nerrf_syscall {
  NERRF_SYS_UNKNOWN = 0,
  NERRF_SYS_OPENAT = 1,
  NERRF_SYS_WRITE = 2,
  NERRF_SYS_RENAME = 3,
  NERRF_SYS_READ = 4,
  NERRF_SYS_UNLINK = 5,
  NERRF_SYS_CHMOD = 6,
};

struct nerrf_event {
  unsigned long long ts_ns;   /* CLOCK_MONOTONIC at capture */
  unsigned int pid;
  unsigned int tid;
  int syscall_id;             /* enum nerrf_syscall */
  int flags;                  /* openat flags (O_RDONLY/O_WRONLY/O_RDWR) */
  long long ret_val;          /* unavailable at sys_enter: 0 */
  unsigned long long bytes;   /* write/read count argument */
  char comm[NERRF_COMM_LEN];
  char path[NERRF_PATH_MAX];
  char new_path[NERRF_PATH_MAX];
};

#define NERRF_RINGBUF_BYTES (1 << 19) /* 512 KiB: 2x the upstream 256 KiB */

#endif /* NERRF_EVENT_ABI_H */
