// UDP multicast peer discovery for dnet_amd.
//
// Native C++ equivalent of the reference's Rust dnet-p2p cdylib
// (reference: lib/dnet-p2p — AsyncDnetP2P.create_instance/async_start/
// async_get_properties/async_set_is_busy; DnetDeviceProperties fields per
// tests/fakes/discovery.py:31-39). The reference broadcasts presence over
// UDP on the LAN; here we announce over UDP multicast (239.192.31.41 by
// default, TTL 1, loop enabled so same-host processes — the MI355X
// one-process-per-GPU layout — see each other), with a background thread
// owning one socket for both announce and listen. Thunderbolt link info is
// replaced by the xGMI probe in parallel/profiler.py; this module only
// does presence + liveness.
//
// Wire format (one datagram, '|'-separated, no JSON dependency):
//   dnetp2p|1|<instance>|<ip>|<http_port>|<shard_port>|<mgr>|<busy>|<gpu>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <arpa/inet.h>
#include <netinet/in.h>
#include <sys/socket.h>
#include <sys/time.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <cstring>
#include <map>
#include <mutex>
#include <sstream>
#include <string>
#include <thread>
#include <vector>

namespace py = pybind11;

namespace {

double now_s() {
    return std::chrono::duration<double>(
               std::chrono::steady_clock::now().time_since_epoch())
        .count();
}

struct Peer {
    std::string instance;
    std::string ip;
    int http_port = 0;
    int shard_port = 0;
    bool is_manager = false;
    bool is_busy = false;
    int gpu_index = -1;
    double last_seen = 0.0;
};

std::vector<std::string> split(const std::string& s, char sep) {
    std::vector<std::string> out;
    std::stringstream ss(s);
    std::string part;
    while (std::getline(ss, part, sep)) out.push_back(part);
    return out;
}

// Outgoing-interface IP for the multicast group (no packet is sent by
// connect() on a UDP socket). Falls back to loopback.
std::string detect_local_ip(const std::string& group, int port) {
    int fd = ::socket(AF_INET, SOCK_DGRAM, 0);
    if (fd < 0) return "127.0.0.1";
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons(static_cast<uint16_t>(port));
    ::inet_pton(AF_INET, group.c_str(), &addr.sin_addr);
    std::string ip = "127.0.0.1";
    if (::connect(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) == 0) {
        sockaddr_in self{};
        socklen_t len = sizeof(self);
        if (::getsockname(fd, reinterpret_cast<sockaddr*>(&self), &len) == 0) {
            char buf[INET_ADDRSTRLEN] = {0};
            if (::inet_ntop(AF_INET, &self.sin_addr, buf, sizeof(buf)))
                ip = buf;
        }
    }
    ::close(fd);
    return ip;
}

}  // namespace

class P2PInstance {
public:
    P2PInstance(std::string instance, int http_port, int shard_port,
                bool is_manager, int gpu_index, std::string group, int port,
                double interval_s, double expire_s)
        : instance_(std::move(instance)),
          http_port_(http_port),
          shard_port_(shard_port),
          is_manager_(is_manager),
          gpu_index_(gpu_index),
          group_(std::move(group)),
          port_(port),
          interval_s_(interval_s),
          expire_s_(expire_s) {
        local_ip_ = detect_local_ip(group_, port_);
    }

    ~P2PInstance() { stop(); }

    void start() {
        if (running_.exchange(true)) return;
        fd_ = ::socket(AF_INET, SOCK_DGRAM, 0);
        if (fd_ < 0) {
            running_ = false;
            throw std::runtime_error("p2p: socket() failed");
        }
        int one = 1;
        ::setsockopt(fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
#ifdef SO_REUSEPORT
        // multicast datagrams are delivered to EVERY socket in the
        // reuseport group — required for several ranks on one host
        ::setsockopt(fd_, SOL_SOCKET, SO_REUSEPORT, &one, sizeof(one));
#endif
        sockaddr_in bind_addr{};
        bind_addr.sin_family = AF_INET;
        bind_addr.sin_addr.s_addr = htonl(INADDR_ANY);
        bind_addr.sin_port = htons(static_cast<uint16_t>(port_));
        if (::bind(fd_, reinterpret_cast<sockaddr*>(&bind_addr),
                   sizeof(bind_addr)) != 0) {
            ::close(fd_);
            fd_ = -1;
            running_ = false;
            throw std::runtime_error("p2p: bind() failed on port " +
                                     std::to_string(port_));
        }
        ip_mreq mreq{};
        ::inet_pton(AF_INET, group_.c_str(), &mreq.imr_multiaddr);
        mreq.imr_interface.s_addr = htonl(INADDR_ANY);
        ::setsockopt(fd_, IPPROTO_IP, IP_ADD_MEMBERSHIP, &mreq, sizeof(mreq));
        unsigned char ttl = 1, loop = 1;
        ::setsockopt(fd_, IPPROTO_IP, IP_MULTICAST_TTL, &ttl, sizeof(ttl));
        ::setsockopt(fd_, IPPROTO_IP, IP_MULTICAST_LOOP, &loop, sizeof(loop));
        timeval tv{};
        tv.tv_usec = 100000;  // 100 ms recv timeout -> responsive shutdown
        ::setsockopt(fd_, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
        thread_ = std::thread([this] { loop_(); });
    }

    void stop() {
        if (!running_.exchange(false)) return;
        if (thread_.joinable()) thread_.join();
        if (fd_ >= 0) ::close(fd_);
        fd_ = -1;
    }

    bool is_running() const { return running_.load(); }
    void set_is_busy(bool b) { busy_.store(b); }
    const std::string& local_ip() const { return local_ip_; }

    // peers map INCLUDING self (self entry is synthesized, not echoed)
    py::dict get_properties() {
        py::dict out;
        out[py::str(instance_)] = props_dict_(self_peer_());
        std::lock_guard<std::mutex> lk(mu_);
        const double now = now_s();
        for (auto it = peers_.begin(); it != peers_.end();) {
            if (now - it->second.last_seen > expire_s_) {
                it = peers_.erase(it);
                continue;
            }
            out[py::str(it->first)] = props_dict_(it->second);
            ++it;
        }
        return out;
    }

private:
    Peer self_peer_() {
        Peer p;
        p.instance = instance_;
        p.ip = local_ip_;
        p.http_port = http_port_;
        p.shard_port = shard_port_;
        p.is_manager = is_manager_;
        p.is_busy = busy_.load();
        p.gpu_index = gpu_index_;
        p.last_seen = now_s();
        return p;
    }

    static py::dict props_dict_(const Peer& p) {
        py::dict d;
        d["instance"] = p.instance;
        d["local_ip"] = p.ip;
        d["server_port"] = p.http_port;
        d["shard_port"] = p.shard_port;
        d["is_manager"] = p.is_manager;
        d["is_busy"] = p.is_busy;
        d["gpu_index"] = p.gpu_index;
        return d;
    }

    std::string announce_() {
        std::stringstream ss;
        ss << "dnetp2p|1|" << instance_ << '|' << local_ip_ << '|'
           << http_port_ << '|' << shard_port_ << '|' << (is_manager_ ? 1 : 0)
           << '|' << (busy_.load() ? 1 : 0) << '|' << gpu_index_;
        return ss.str();
    }

    void handle_(const std::string& msg) {
        auto f = split(msg, '|');
        if (f.size() < 9 || f[0] != "dnetp2p" || f[1] != "1") return;
        if (f[2] == instance_) return;  // own loopback echo
        Peer p;
        try {
            p.instance = f[2];
            p.ip = f[3];
            p.http_port = std::stoi(f[4]);
            p.shard_port = std::stoi(f[5]);
            p.is_manager = f[6] == "1";
            p.is_busy = f[7] == "1";
            p.gpu_index = std::stoi(f[8]);
        } catch (const std::exception&) {
            return;
        }
        p.last_seen = now_s();
        std::lock_guard<std::mutex> lk(mu_);
        peers_[p.instance] = p;
    }

    void loop_() {
        sockaddr_in dst{};
        dst.sin_family = AF_INET;
        dst.sin_port = htons(static_cast<uint16_t>(port_));
        ::inet_pton(AF_INET, group_.c_str(), &dst.sin_addr);
        double last_tx = 0.0;
        char buf[1024];
        while (running_.load()) {
            const double now = now_s();
            if (now - last_tx >= interval_s_) {
                const std::string msg = announce_();
                ::sendto(fd_, msg.data(), msg.size(), 0,
                         reinterpret_cast<sockaddr*>(&dst), sizeof(dst));
                last_tx = now;
            }
            const ssize_t n = ::recv(fd_, buf, sizeof(buf) - 1, 0);
            if (n > 0) handle_(std::string(buf, static_cast<size_t>(n)));
        }
    }

    std::string instance_;
    int http_port_;
    int shard_port_;
    bool is_manager_;
    int gpu_index_;
    std::string group_;
    int port_;
    double interval_s_;
    double expire_s_;
    std::string local_ip_;
    std::atomic<bool> running_{false};
    std::atomic<bool> busy_{false};
    int fd_ = -1;
    std::thread thread_;
    std::mutex mu_;
    std::map<std::string, Peer> peers_;
};

PYBIND11_MODULE(_p2p, m) {
    m.doc() = "UDP multicast peer discovery (native)";
    py::class_<P2PInstance>(m, "P2PInstance")
        .def(py::init<std::string, int, int, bool, int, std::string, int,
                      double, double>(),
             py::arg("instance"), py::arg("http_port"), py::arg("shard_port"),
             py::arg("is_manager") = false, py::arg("gpu_index") = -1,
             py::arg("group") = "239.192.31.41", py::arg("port") = 52525,
             py::arg("interval_s") = 0.5, py::arg("expire_s") = 5.0)
        .def("start", &P2PInstance::start,
             py::call_guard<py::gil_scoped_release>())
        .def("stop", &P2PInstance::stop,
             py::call_guard<py::gil_scoped_release>())
        .def("is_running", &P2PInstance::is_running)
        .def("set_is_busy", &P2PInstance::set_is_busy)
        .def("get_properties", &P2PInstance::get_properties)
        .def_property_readonly("local_ip", &P2PInstance::local_ip);
}
