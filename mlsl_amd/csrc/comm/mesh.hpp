// Full-mesh nonblocking TCP data plane for the host transport.
//
// MI355X-native re-design of the reference's eplib shared-memory command
// queues (eplib/cqueue.c): here the progress thread owns all sockets and
// pumps framed messages; collectives are resumable schedules advanced against
// this mesh. Only the progress thread touches Channel state (SPSC by
// construction, like the reference's lock-free cqueue).
//
// Frame: [u64 tag][u64 len][payload]. At most one in-flight outgoing frame
// per channel; receivers demux by tag into posted buffers, with an
// unexpected-message queue for early arrivals.
#pragma once

#include <cstdint>
#include <deque>
#include <memory>
#include <unordered_map>
#include <vector>

namespace mlsl {

class Bootstrap;

struct PendingRecv {
    uint8_t* buf = nullptr;
    size_t len = 0;
    size_t got = 0;
    bool* done_flag = nullptr;   // set true on completion
};

struct UnexpectedMsg {
    std::vector<uint8_t> data;
    size_t got = 0;
    bool complete = false;
};

class Channel {
  public:
    explicit Channel(int fd);
    ~Channel();

    // Begin sending a frame if the socket is idle. Returns false if another
    // frame is in flight (caller retries on a later progress tick).
    bool StartSend(uint64_t tag, const void* buf, size_t len, bool* done_flag);
    // Post a receive for `tag`; completes immediately if the message already
    // arrived unexpectedly.
    void PostRecv(uint64_t tag, void* buf, size_t len, bool* done_flag);
    // Pump the socket: flush outgoing frame bytes, drain incoming bytes.
    void Progress();

    bool SendIdle() const { return !send_active_; }

  private:
    void ProgressSend();
    void ProgressRecv();

    int fd_;
    bool peer_closed_ = false;
    // --- send side ---
    bool send_active_ = false;
    uint8_t send_hdr_[16];
    size_t send_hdr_sent_ = 0;
    const uint8_t* send_payload_ = nullptr;
    size_t send_len_ = 0;
    size_t send_sent_ = 0;
    bool* send_done_flag_ = nullptr;
    // --- recv side ---
    uint8_t recv_hdr_[16];
    size_t recv_hdr_got_ = 0;
    bool recv_in_msg_ = false;
    uint64_t recv_tag_ = 0;
    uint64_t recv_len_ = 0;
    PendingRecv* recv_cur_ = nullptr;        // posted target, or
    UnexpectedMsg* recv_unexp_ = nullptr;    // unexpected buffer
    std::unordered_map<uint64_t, PendingRecv> posted_;
    // multimap: several in-flight messages may carry the same tag (e.g. a
    // buggy or replayed sender) — aliasing them corrupted stream framing
    std::unordered_multimap<uint64_t, UnexpectedMsg> unexpected_;
};

class Mesh {
  public:
    // Establish the full mesh using the bootstrap for address exchange.
    explicit Mesh(Bootstrap& boot);
    ~Mesh();

    int Rank() const { return rank_; }
    int Size() const { return size_; }

    bool StartSend(int peer, uint64_t tag, const void* buf, size_t len, bool* done);
    void PostRecv(int peer, uint64_t tag, void* buf, size_t len, bool* done);
    void Progress();

  private:
    int rank_ = 0, size_ = 1;
    int listen_fd_ = -1;
    std::vector<std::unique_ptr<Channel>> chans_;  // index by peer world rank
};

}  // namespace mlsl
