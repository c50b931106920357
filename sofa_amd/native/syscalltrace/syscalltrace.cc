// sofa-syscalltrace — ptrace-based syscall tracer (strace replacement).
//
// The reference shells out to `strace -q -T -tt -f` (cyliustack/sofa
// bin/sofa_record.py:336-337) and re-parses its text (bin/sofa_preprocess.py:
// 1623-1704); this image ships no strace, so the tracer is native: launch the
// target under PTRACE_TRACEME, follow forks/clones/execs, stamp every
// syscall enter/exit with CLOCK_MONOTONIC_RAW, and write fixed 40-byte
// records for vectorized numpy parsing (sofa_amd/preprocess/strace.py).
//
// Usage: sofa-syscalltrace -o out.sst -- <cmd> [args...]
// x86-64 only (orig_rax/rax register ABI).

#include <sys/ptrace.h>
#include <sys/user.h>
#include <sys/wait.h>
#include <unistd.h>
#include <signal.h>

#include <cerrno>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <ctime>
#include <unordered_map>
#include <vector>

namespace {

constexpr uint32_t kMagic = 0x31545353;  // "SST1"

struct FileHeader {
  uint32_t magic;
  uint32_t version;
  uint64_t realtime_ns;
  uint64_t monotonic_raw_ns;
  uint64_t reserved;
};

struct SysRec {
  uint64_t t_enter_ns;  // CLOCK_MONOTONIC_RAW
  uint64_t duration_ns;
  uint32_t tid;
  uint32_t sysno;
  int64_t ret;
};
static_assert(sizeof(SysRec) == 32, "SysRec is 32 bytes");

uint64_t now_ns(clockid_t c) {
  struct timespec ts;
  clock_gettime(c, &ts);
  return uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
}

struct Pending {
  uint64_t t_enter;
  uint32_t sysno;
  bool in_syscall = false;
};

}  // namespace

int main(int argc, char** argv) {
  const char* out_path = nullptr;
  int cmd_start = -1;
  for (int i = 1; i < argc; i++) {
    if (!strcmp(argv[i], "-o") && i + 1 < argc) out_path = argv[++i];
    else if (!strcmp(argv[i], "--")) { cmd_start = i + 1; break; }
  }
  if (!out_path || cmd_start < 0 || cmd_start >= argc) {
    fprintf(stderr, "usage: %s -o out.sst -- cmd [args...]\n", argv[0]);
    return 2;
  }

  pid_t child = fork();
  if (child == 0) {
    ptrace(PTRACE_TRACEME, 0, nullptr, nullptr);
    raise(SIGSTOP);
    execvp(argv[cmd_start], &argv[cmd_start]);
    perror("execvp");
    _exit(127);
  }

  FILE* f = fopen(out_path, "wb");
  if (!f) { perror("fopen"); return 1; }
  FileHeader hdr{};
  hdr.magic = kMagic;
  hdr.version = 1;
  hdr.realtime_ns = now_ns(CLOCK_REALTIME);
  hdr.monotonic_raw_ns = now_ns(CLOCK_MONOTONIC_RAW);
  fwrite(&hdr, sizeof(hdr), 1, f);

  int status = 0;
  waitpid(child, &status, 0);  // SIGSTOP from TRACEME child
  long opts = PTRACE_O_TRACESYSGOOD | PTRACE_O_TRACECLONE | PTRACE_O_TRACEFORK |
              PTRACE_O_TRACEVFORK | PTRACE_O_TRACEEXEC | PTRACE_O_EXITKILL;
  ptrace(PTRACE_SETOPTIONS, child, nullptr, (void*) opts);
  ptrace(PTRACE_SYSCALL, child, nullptr, nullptr);

  std::unordered_map<pid_t, Pending> pending;
  std::vector<SysRec> batch;
  batch.reserve(4096);
  int live = 1;

  while (live > 0) {
    pid_t tid = waitpid(-1, &status, __WALL);
    if (tid < 0) {
      if (errno == ECHILD) break;
      continue;
    }
    if (WIFEXITED(status) || WIFSIGNALED(status)) {
      pending.erase(tid);
      live--;
      continue;
    }
    if (!WIFSTOPPED(status)) continue;
    int sig = WSTOPSIG(status);
    int event = (status >> 16) & 0xFF;
    if (event == PTRACE_EVENT_CLONE || event == PTRACE_EVENT_FORK ||
        event == PTRACE_EVENT_VFORK) {
      live++;  // new tracee inherits options and will report
      ptrace(PTRACE_SYSCALL, tid, nullptr, nullptr);
      continue;
    }
    if (sig == (SIGTRAP | 0x80)) {
      // syscall-stop
      auto& p = pending[tid];
      struct user_regs_struct regs;
      if (ptrace(PTRACE_GETREGS, tid, nullptr, &regs) == 0) {
        if (!p.in_syscall) {
          p.in_syscall = true;
          p.sysno = (uint32_t) regs.orig_rax;
          p.t_enter = now_ns(CLOCK_MONOTONIC_RAW);
        } else {
          p.in_syscall = false;
          SysRec r{};
          r.t_enter_ns = p.t_enter;
          r.duration_ns = now_ns(CLOCK_MONOTONIC_RAW) - p.t_enter;
          r.tid = (uint32_t) tid;
          r.sysno = p.sysno;
          r.ret = (int64_t) regs.rax;
          batch.push_back(r);
          if (batch.size() >= 4096) {
            fwrite(batch.data(), sizeof(SysRec), batch.size(), f);
            batch.clear();
          }
        }
      }
      ptrace(PTRACE_SYSCALL, tid, nullptr, nullptr);
    } else if (sig == SIGTRAP || event != 0) {
      ptrace(PTRACE_SYSCALL, tid, nullptr, nullptr);
    } else if (sig == SIGSTOP && pending.find(tid) == pending.end()) {
      // a new tracee's initial SIGSTOP: suppress it (re-injecting would
      // group-stop the thread), start syscall-tracing it
      pending[tid];  // mark seen
      ptrace(PTRACE_SYSCALL, tid, nullptr, nullptr);
    } else {
      // deliver the real signal
      ptrace(PTRACE_SYSCALL, tid, nullptr, (void*) (long) sig);
    }
  }
  if (!batch.empty()) fwrite(batch.data(), sizeof(SysRec), batch.size(), f);
  fclose(f);
  return 0;
}
