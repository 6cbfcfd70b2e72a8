// MPI-free bootstrap: a tiny TCP rendezvous (star on world rank 0) used to
// (a) form the world, (b) allgather small control blobs (peer addresses,
// group colors, RCCL unique ids). Replaces the reference's dependency on a
// bundled MPI runtime (mpirt/, PMPI_Init) — per BASELINE.json north star the
// rebuild has no MPI runtime dependency.
//
// Rank/size/address come from the standard torchrun-style env
// (RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT, MLSL_* overrides); rendezvous
// port defaults to MASTER_PORT+1 so it can coexist with a torch.distributed
// store in the same job.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace mlsl {

class Bootstrap {
  public:
    // rank/size < 0 → read from env. Blocks until the world is connected.
    Bootstrap(int rank, int size);
    ~Bootstrap();

    int Rank() const { return rank_; }
    int Size() const { return size_; }

    // Fixed-size allgather: every rank contributes `len` bytes; `out` gets
    // size*len bytes ordered by rank. Collective over the world.
    void Allgather(const void* mine, size_t len, void* out);
    void Barrier();

    static int EnvRank();
    static int EnvSize();

  private:
    int rank_ = 0;
    int size_ = 1;
    int root_listen_ = -1;            // root only
    std::vector<int> socks_;          // root: per-rank sockets; others: [0]=root
};

// Low-level helpers shared with the mesh.
int TcpListen(const char* host, int port, int backlog, int* bound_port);
int TcpConnectRetry(const std::string& host, int port, int timeout_sec);
void TcpSendAll(int fd, const void* buf, size_t len);
void TcpRecvAll(int fd, void* buf, size_t len);
void TcpSetNonBlocking(int fd, bool nb);
void TcpSetNoDelay(int fd);

}  // namespace mlsl
