// bootstrap.cpp — multi-node RCCL bootstrap.
//
// The reference's Inspur delta is a 46-line MPI shim (clusters.cpp) whose
// only data-plane job is broadcasting the ncclUniqueId so ONE communicator
// spans every GPU of every node (parallel.cpp:42-45,166-169).  Single-node
// needs neither MPI nor a network; multi-node needs exactly a 128-byte
// exchange — done here over plain TCP: the global rank 0 serves the id on
// MASTER_PORT, every other rank connects and reads it.  RCCL itself
// handles all transport after init (sockets/IB per its own env).
//
// The `caffe` CLI picks this up from the environment (tools/caffe_main):
//   CAFFE_NNODES    total nodes (default 1 — no network touched)
//   CAFFE_NODE_RANK this node's rank (0-based)
//   MASTER_ADDR / MASTER_PORT   rank-0 endpoint (torchrun's convention)
// Global rank = node_rank * gpus_per_node + local_rank, exactly the
// reference's (node_rank * nranks + rank) mapping (parallel.cpp:166-169).
#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>

#include "core.hpp"

namespace camd {

// serve `payload` (len bytes) to exactly `nclients` connections, then
// close.  Returns 0 on success.
int uid_serve(const void* payload, int len, int port, int nclients) {
  const int srv = socket(AF_INET, SOCK_STREAM, 0);
  if (srv < 0) return -1;
  int one = 1;
  setsockopt(srv, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
  sockaddr_in addr{};
  addr.sin_family = AF_INET;
  addr.sin_addr.s_addr = htonl(INADDR_ANY);
  addr.sin_port = htons((uint16_t)port);
  if (bind(srv, (sockaddr*)&addr, sizeof(addr)) != 0 ||
      listen(srv, nclients + 8) != 0) {
    close(srv);
    return -1;
  }
  for (int i = 0; i < nclients; ++i) {
    const int c = accept(srv, nullptr, nullptr);
    if (c < 0) {
      close(srv);
      return -1;
    }
    const char* p = (const char*)payload;
    int left = len;
    while (left > 0) {
      const ssize_t w = write(c, p, left);
      if (w <= 0) break;
      p += w;
      left -= (int)w;
    }
    close(c);
    if (left != 0) {
      close(srv);
      return -1;
    }
  }
  close(srv);
  return 0;
}

// fetch len bytes from host:port, retrying connection for up to
// timeout_s seconds (the server may not be up yet).  Returns 0 on
// success.
int uid_fetch(void* out, int len, const char* host, int port,
              int timeout_s) {
  addrinfo hints{}, *res = nullptr;
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  char portstr[16];
  snprintf(portstr, sizeof(portstr), "%d", port);
  if (getaddrinfo(host, portstr, &hints, &res) != 0 || !res) return -1;
  int rc = -1;
  for (int attempt = 0; attempt < timeout_s * 10; ++attempt) {
    const int s = socket(AF_INET, SOCK_STREAM, 0);
    if (s < 0) break;
    if (connect(s, res->ai_addr, res->ai_addrlen) == 0) {
      char* p = (char*)out;
      int left = len;
      while (left > 0) {
        const ssize_t r = read(s, p, left);
        if (r <= 0) break;
        p += r;
        left -= (int)r;
      }
      close(s);
      if (left == 0) {
        rc = 0;
        break;
      }
      break;  // short read from a live server: fail loudly
    }
    close(s);
    usleep(100000);  // server not up yet — retry
  }
  freeaddrinfo(res);
  return rc;
}

}  // namespace camd
