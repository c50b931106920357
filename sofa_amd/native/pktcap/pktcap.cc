// sofa-pktcap — native packet-header capture on AF_PACKET.
//
// Replaces the reference's tcpdump dependency (cyliustack/sofa
// bin/sofa_record.py:291-298 + pcap text re-parse in
// bin/sofa_preprocess.py:1188-1231): the image ships no tcpdump, and headers
// are all the profiler needs.  A cooked AF_PACKET socket captures every
// interface; only IPv4 TCP/UDP headers are decoded; 32-byte fixed records are
// written for vectorized numpy parsing (sofa_amd/preprocess/net.py).

#include <arpa/inet.h>
#include <linux/if_ether.h>
#include <linux/if_packet.h>
#include <netinet/ip.h>
#include <netinet/tcp.h>
#include <netinet/udp.h>
#include <signal.h>
#include <sys/prctl.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cerrno>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <ctime>
#include <string_view>

namespace {

constexpr uint32_t kMagic = 0x31435053;  // "SPC1"

struct FileHeader {
  uint32_t magic;
  uint32_t version;
  uint64_t realtime_ns;
  uint64_t monotonic_raw_ns;
  uint64_t reserved;
};

struct PktRec {
  uint64_t time_ns;  // CLOCK_MONOTONIC_RAW at recv
  uint32_t src_ip;   // host byte order
  uint32_t dst_ip;
  uint16_t sport;
  uint16_t dport;
  uint32_t len;      // IP total length
  uint8_t proto;     // IPPROTO_TCP / IPPROTO_UDP
  uint8_t dir;       // PACKET_OUTGOING=1, else 0
  uint8_t pad[2];
};
static_assert(sizeof(PktRec) == 32, "PktRec must be 32 bytes");

volatile sig_atomic_t g_stop = 0;
void on_signal(int) { g_stop = 1; }

uint64_t mono_raw_ns() {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC_RAW, &ts);
  return uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
}

}  // namespace

int main(int argc, char** argv) {
  const char* out_path = nullptr;
  uint64_t max_mb = 256;
  for (int i = 1; i < argc; i++) {
    std::string_view a = argv[i];
    if (a == "-o" && i + 1 < argc) out_path = argv[++i];
    else if (a == "--max-mb" && i + 1 < argc) max_mb = strtoull(argv[++i], nullptr, 10);
  }
  if (!out_path) {
    fprintf(stderr, "usage: %s -o out.bin [--max-mb N]\n", argv[0]);
    return 2;
  }
  signal(SIGTERM, on_signal);
  signal(SIGINT, on_signal);
  prctl(PR_SET_PDEATHSIG, SIGTERM);

  int sock = socket(AF_PACKET, SOCK_DGRAM, htons(ETH_P_IP));
  if (sock < 0) {
    fprintf(stderr, "sofa-pktcap: socket: %s\n", strerror(errno));
    return 1;
  }
  struct timeval tv = {0, 200000};
  setsockopt(sock, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));

  FILE* f = fopen(out_path, "wb");
  if (!f) {
    perror("sofa-pktcap: fopen");
    return 1;
  }
  FileHeader hdr{};
  hdr.magic = kMagic;
  hdr.version = 1;
  struct timespec ts;
  clock_gettime(CLOCK_REALTIME, &ts);
  hdr.realtime_ns = uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
  hdr.monotonic_raw_ns = mono_raw_ns();
  fwrite(&hdr, sizeof(hdr), 1, f);

  uint64_t written = sizeof(hdr);
  const uint64_t max_bytes = max_mb << 20;
  uint8_t buf[2048];
  PktRec batch[1024];
  size_t n_batch = 0;

  while (!g_stop && written + n_batch * sizeof(PktRec) < max_bytes) {
    struct sockaddr_ll addr;
    socklen_t alen = sizeof(addr);
    ssize_t n = recvfrom(sock, buf, sizeof(buf), 0, (struct sockaddr*) &addr, &alen);
    if (n < 0) {
      if (errno == EAGAIN || errno == EWOULDBLOCK || errno == EINTR) {
        if (n_batch) {
          fwrite(batch, sizeof(PktRec), n_batch, f);
          written += n_batch * sizeof(PktRec);
          n_batch = 0;
        }
        continue;
      }
      break;
    }
    if ((size_t) n < sizeof(struct iphdr)) continue;
    auto* ip = reinterpret_cast<struct iphdr*>(buf);
    if (ip->version != 4) continue;
    size_t ihl = size_t(ip->ihl) * 4;
    PktRec r{};
    r.time_ns = mono_raw_ns();
    r.src_ip = ntohl(ip->saddr);
    r.dst_ip = ntohl(ip->daddr);
    r.len = ntohs(ip->tot_len);
    r.proto = ip->protocol;
    r.dir = addr.sll_pkttype == PACKET_OUTGOING ? 1 : 0;
    if (ip->protocol == IPPROTO_TCP && (size_t) n >= ihl + sizeof(struct tcphdr)) {
      auto* tcp = reinterpret_cast<struct tcphdr*>(buf + ihl);
      r.sport = ntohs(tcp->source);
      r.dport = ntohs(tcp->dest);
    } else if (ip->protocol == IPPROTO_UDP && (size_t) n >= ihl + sizeof(struct udphdr)) {
      auto* udp = reinterpret_cast<struct udphdr*>(buf + ihl);
      r.sport = ntohs(udp->source);
      r.dport = ntohs(udp->dest);
    }
    batch[n_batch++] = r;
    if (n_batch == 1024) {
      fwrite(batch, sizeof(PktRec), n_batch, f);
      written += n_batch * sizeof(PktRec);
      n_batch = 0;
    }
  }
  if (n_batch) fwrite(batch, sizeof(PktRec), n_batch, f);
  fclose(f);
  close(sock);
  return 0;
}
