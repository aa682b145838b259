/* Probe why BPF_PROG_LOAD / PROG_ATTACH fail on a box.
 * Build: gcc -O1 -o /tmp/bpf_probe scripts/bpf_probe.c */
#define _GNU_SOURCE
#include <errno.h>
#include <fcntl.h>
#include <stdint.h>
#include <stdio.h>
#include <string.h>
#include <sys/syscall.h>
#include <unistd.h>

struct insn { uint8_t code, regs; int16_t off; int32_t imm; };

int main(int argc, char** argv) {
  const char* cgdir = argc > 1 ? argv[1] : NULL;
  struct insn prog[2] = {
    {0xb7, 0, 0, 1},  /* mov r0, 1 (allow all) */
    {0x95, 0, 0, 0},  /* exit */
  };
  union { struct { uint32_t prog_type, insn_cnt; uint64_t insns, license;
                   uint32_t log_level, log_size; uint64_t log_buf;
                   uint32_t kern_version, prog_flags; } load;
          struct { uint32_t target_fd, attach_bpf_fd, attach_type, attach_flags; } attach;
          char pad[128]; } attr;
  static char logbuf[4096];
  memset(&attr, 0, sizeof attr);
  attr.load.prog_type = 15; /* CGROUP_DEVICE */
  attr.load.insn_cnt = 2;
  attr.load.insns = (uint64_t)(uintptr_t)prog;
  attr.load.license = (uint64_t)(uintptr_t)"GPL";
  attr.load.log_level = 1;
  attr.load.log_size = sizeof logbuf;
  attr.load.log_buf = (uint64_t)(uintptr_t)logbuf;
  int fd = syscall(SYS_bpf, 5, &attr, sizeof attr);
  printf("PROG_LOAD: fd=%d errno=%d (%s)\nverifier: %s\n", fd, errno,
         strerror(errno), logbuf);
  if (fd < 0) return 1;
  if (cgdir) {
    int cg = open(cgdir, O_DIRECTORY | O_RDONLY);
    printf("open(%s): %d errno=%d\n", cgdir, cg, errno);
    if (cg >= 0) {
      memset(&attr, 0, sizeof attr);
      attr.attach.target_fd = cg;
      attr.attach.attach_bpf_fd = fd;
      attr.attach.attach_type = 6;  /* BPF_CGROUP_DEVICE */
      attr.attach.attach_flags = 1; /* ALLOW_OVERRIDE */
      int rc = syscall(SYS_bpf, 8, &attr, sizeof attr);
      printf("PROG_ATTACH: rc=%d errno=%d (%s)\n", rc, errno, strerror(errno));
    }
  }
  return 0;
}
