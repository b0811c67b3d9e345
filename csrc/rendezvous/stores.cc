#include "rendezvous/stores.h"

#include <arpa/inet.h>
#include <fcntl.h>
#include <sys/file.h>
#include <netdb.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <sys/stat.h>
#include <unistd.h>

#include <atomic>
#include <cstring>
#include <fstream>
#include <vector>

#include "common/logging.h"

namespace glooamd {

// ---------------------------------------------------------------------------
// HashStore
// ---------------------------------------------------------------------------

void HashStore::set(const std::string& key, const std::vector<char>& data) {
  std::lock_guard<std::mutex> lock(mu_);
  map_[key] = data;
  cv_.notify_all();
}

std::vector<char> HashStore::get(const std::string& key) {
  std::unique_lock<std::mutex> lock(mu_);
  if (!cv_.wait_for(lock, kDefaultTimeout, [&] { return map_.count(key); })) {
    throw TimeoutException("HashStore::get timeout for key " + key);
  }
  return map_[key];
}

void HashStore::wait(
    const std::vector<std::string>& keys,
    const std::chrono::milliseconds& timeout) {
  std::unique_lock<std::mutex> lock(mu_);
  auto pred = [&] {
    for (const auto& k : keys) {
      if (map_.count(k) == 0) {
        return false;
      }
    }
    return true;
  };
  if (timeout.count() < 0) {
    cv_.wait(lock, pred); // negative timeout = wait forever
  } else if (!cv_.wait_for(lock, timeout, pred)) {
    throw TimeoutException("HashStore::wait timeout");
  }
}

void HashStore::append(const std::string& key, const std::vector<char>& data) {
  std::lock_guard<std::mutex> lock(mu_);
  auto& v = map_[key];
  v.insert(v.end(), data.begin(), data.end());
  cv_.notify_all();
}

int64_t HashStore::add(const std::string& key, int64_t delta) {
  std::lock_guard<std::mutex> lock(mu_);
  auto& v = map_[key];
  int64_t cur = v.empty() ? 0 : strtoll(std::string(v.begin(), v.end()).c_str(), nullptr, 10);
  cur += delta;
  std::string s = std::to_string(cur);
  v.assign(s.begin(), s.end());
  cv_.notify_all();
  return cur;
}

// ---------------------------------------------------------------------------
// FileStore — tmp-file + rename for atomic publication; 10ms poll on wait.
// ---------------------------------------------------------------------------

FileStore::FileStore(const std::string& path) : basePath_(path) {
  mkdir(path.c_str(), 0777); // ok if it exists
}

static std::string encodeKey(const std::string& key) {
  // keys may contain '/'; hex-escape everything non-alnum
  std::string out;
  for (char c : key) {
    if (isalnum(static_cast<unsigned char>(c)) || c == '-' || c == '_') {
      out.push_back(c);
    } else {
      char buf[8];
      snprintf(buf, sizeof(buf), ".%02x", static_cast<unsigned char>(c));
      out += buf;
    }
  }
  return out;
}

std::string FileStore::objectPath(const std::string& key) const {
  return basePath_ + "/" + encodeKey(key);
}

void FileStore::set(const std::string& key, const std::vector<char>& data) {
  static std::atomic<uint64_t> counter{0};
  std::string tmp = basePath_ + "/.tmp." + std::to_string(getpid()) + "." +
      std::to_string(counter.fetch_add(1));
  {
    std::ofstream f(tmp, std::ios::binary | std::ios::trunc);
    GA_ENFORCE(f.good(), "FileStore: cannot write ", tmp);
    f.write(data.data(), data.size());
  }
  GA_ENFORCE_EQ(
      rename(tmp.c_str(), objectPath(key).c_str()),
      0,
      "FileStore rename: ",
      strerror(errno));
}

std::vector<char> FileStore::get(const std::string& key) {
  wait({key}, kDefaultTimeout);
  std::ifstream f(objectPath(key), std::ios::binary);
  GA_ENFORCE(f.good(), "FileStore: cannot read key ", key);
  return std::vector<char>(
      std::istreambuf_iterator<char>(f), std::istreambuf_iterator<char>());
}

namespace {
// RAII flock on a per-store lock file: cross-process atomicity for the
// read-modify-write v2 ops.
class FileLock {
 public:
  explicit FileLock(const std::string& path) {
    fd_ = open(path.c_str(), O_CREAT | O_RDWR | O_CLOEXEC, 0666);
    GA_ENFORCE_GE(fd_, 0, "FileStore lock open: ", strerror(errno));
    GA_ENFORCE_EQ(flock(fd_, LOCK_EX), 0, "flock: ", strerror(errno));
  }
  ~FileLock() {
    if (fd_ >= 0) {
      flock(fd_, LOCK_UN);
      close(fd_);
    }
  }

 private:
  int fd_{-1};
};

std::vector<char> readFileIfExists(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  if (!f.good()) {
    return {};
  }
  return std::vector<char>(
      std::istreambuf_iterator<char>(f), std::istreambuf_iterator<char>());
}
} // namespace

void FileStore::append(const std::string& key, const std::vector<char>& data) {
  FileLock lock(basePath_ + "/.lock");
  auto cur = readFileIfExists(objectPath(key));
  cur.insert(cur.end(), data.begin(), data.end());
  set(key, cur); // tmp+rename keeps readers atomic too
}

int64_t FileStore::add(const std::string& key, int64_t delta) {
  FileLock lock(basePath_ + "/.lock");
  auto cur = readFileIfExists(objectPath(key));
  int64_t v = cur.empty()
      ? 0
      : strtoll(std::string(cur.begin(), cur.end()).c_str(), nullptr, 10);
  v += delta;
  std::string s = std::to_string(v);
  set(key, std::vector<char>(s.begin(), s.end()));
  return v;
}

bool FileStore::check(const std::vector<std::string>& keys) const {
  for (const auto& k : keys) {
    struct stat st;
    if (stat(objectPath(k).c_str(), &st) != 0) {
      return false;
    }
  }
  return true;
}

void FileStore::wait(
    const std::vector<std::string>& keys,
    const std::chrono::milliseconds& timeout) {
  auto deadline = std::chrono::steady_clock::now() + timeout;
  while (!check(keys)) {
    if (timeout.count() >= 0 && std::chrono::steady_clock::now() > deadline) {
      throw TimeoutException("FileStore::wait timeout");
    }
    usleep(10 * 1000);
  }
}

// ---------------------------------------------------------------------------
// PrefixStore
// ---------------------------------------------------------------------------

PrefixStore::PrefixStore(const std::string& prefix, std::shared_ptr<IStore> s)
    : prefix_(prefix), store_(std::move(s)) {}

void PrefixStore::set(const std::string& key, const std::vector<char>& data) {
  store_->set(prefix_ + "/" + key, data);
}

std::vector<char> PrefixStore::get(const std::string& key) {
  return store_->get(prefix_ + "/" + key);
}

void PrefixStore::wait(
    const std::vector<std::string>& keys,
    const std::chrono::milliseconds& timeout) {
  std::vector<std::string> prefixed;
  prefixed.reserve(keys.size());
  for (const auto& k : keys) {
    prefixed.push_back(prefix_ + "/" + k);
  }
  store_->wait(prefixed, timeout);
}

void PrefixStore::append(
    const std::string& key,
    const std::vector<char>& data) {
  store_->append(prefix_ + "/" + key, data);
}

int64_t PrefixStore::add(const std::string& key, int64_t delta) {
  return store_->add(prefix_ + "/" + key, delta);
}

// ---------------------------------------------------------------------------
// TcpStore
// ---------------------------------------------------------------------------

namespace {

void writeAll(int fd, const void* buf, size_t n) {
  const char* p = static_cast<const char*>(buf);
  size_t done = 0;
  while (done < n) {
    ssize_t rv = ::write(fd, p + done, n - done);
    if (rv < 0 && errno == EINTR) {
      continue;
    }
    GA_ENFORCE_GT(rv, 0, "TcpStore write: ", strerror(errno));
    done += rv;
  }
}

bool readAll(int fd, void* buf, size_t n) {
  char* p = static_cast<char*>(buf);
  size_t done = 0;
  while (done < n) {
    ssize_t rv = ::read(fd, p + done, n - done);
    if (rv == 0) {
      return false; // EOF
    }
    if (rv < 0 && errno == EINTR) {
      continue;
    }
    GA_ENFORCE_GT(rv, 0, "TcpStore read: ", strerror(errno));
    done += rv;
  }
  return true;
}

void writeString(int fd, const std::string& s) {
  uint32_t len = s.size();
  writeAll(fd, &len, 4);
  writeAll(fd, s.data(), s.size());
}

} // namespace

// The server: accepts clients; each client handled by one thread (client
// count == world size: cheap). GET blocks on a condvar until the key is set.
class TcpStore::Server {
 public:
  explicit Server(int port) {
    listenFd_ = socket(AF_INET, SOCK_STREAM | SOCK_CLOEXEC, 0);
    GA_ENFORCE_GE(listenFd_, 0);
    int on = 1;
    setsockopt(listenFd_, SOL_SOCKET, SO_REUSEADDR, &on, sizeof(on));
    struct sockaddr_in addr;
    std::memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(INADDR_ANY);
    addr.sin_port = htons(port);
    GA_ENFORCE_EQ(
        bind(listenFd_, reinterpret_cast<struct sockaddr*>(&addr), sizeof(addr)),
        0,
        "TcpStore bind port ",
        port,
        ": ",
        strerror(errno));
    GA_ENFORCE_EQ(listen(listenFd_, 128), 0);
    acceptThread_ = std::thread([this] { acceptLoop(); });
  }

  ~Server() {
    done_ = true;
    shutdown(listenFd_, SHUT_RDWR);
    close(listenFd_);
    if (acceptThread_.joinable()) {
      acceptThread_.join();
    }
    {
      std::lock_guard<std::mutex> lock(mu_);
      for (int fd : clientFds_) {
        shutdown(fd, SHUT_RDWR);
      }
      cv_.notify_all();
    }
    for (auto& t : clientThreads_) {
      if (t.joinable()) {
        t.join();
      }
    }
  }

 private:
  void acceptLoop() {
    for (;;) {
      int fd = accept4(listenFd_, nullptr, nullptr, SOCK_CLOEXEC);
      if (fd < 0) {
        if (done_) {
          return;
        }
        if (errno == EINTR) {
          continue;
        }
        return;
      }
      int on = 1;
      setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &on, sizeof(on));
      std::lock_guard<std::mutex> lock(mu_);
      clientFds_.push_back(fd);
      clientThreads_.emplace_back([this, fd] { clientLoop(fd); });
    }
  }

  void clientLoop(int fd) {
    for (;;) {
      uint8_t op;
      if (!readAll(fd, &op, 1)) {
        break;
      }
      uint32_t klen;
      if (!readAll(fd, &klen, 4)) {
        break;
      }
      std::string key(klen, '\0');
      if (!readAll(fd, &key[0], klen)) {
        break;
      }
      if (op == 'S' || op == 'A') {
        uint32_t vlen;
        if (!readAll(fd, &vlen, 4)) {
          break;
        }
        std::vector<char> val(vlen);
        if (vlen > 0 && !readAll(fd, val.data(), vlen)) {
          break;
        }
        std::lock_guard<std::mutex> lock(mu_);
        if (op == 'S') {
          map_[key] = std::move(val);
        } else {
          auto& cur = map_[key];
          cur.insert(cur.end(), val.begin(), val.end());
        }
        cv_.notify_all();
      } else if (op == 'I') {
        int64_t delta;
        if (!readAll(fd, &delta, 8)) {
          break;
        }
        int64_t result;
        {
          std::lock_guard<std::mutex> lock(mu_);
          auto& cur = map_[key];
          int64_t v = cur.empty()
              ? 0
              : strtoll(
                    std::string(cur.begin(), cur.end()).c_str(), nullptr, 10);
          v += delta;
          std::string sv = std::to_string(v);
          cur.assign(sv.begin(), sv.end());
          result = v;
          cv_.notify_all();
        }
        writeAll(fd, &result, 8);
      } else if (op == 'C') {
        uint8_t exists;
        {
          std::lock_guard<std::mutex> lock(mu_);
          exists = map_.count(key) > 0 ? 1 : 0;
        }
        writeAll(fd, &exists, 1);
      } else if (op == 'G' || op == 'W') {
        std::unique_lock<std::mutex> lock(mu_);
        cv_.wait(lock, [&] { return map_.count(key) > 0 || done_.load(); });
        if (done_) {
          break;
        }
        if (op == 'G') {
          auto& val = map_[key];
          uint32_t vlen = val.size();
          lock.unlock();
          writeAll(fd, &vlen, 4);
          if (vlen > 0) {
            writeAll(fd, val.data(), vlen);
          }
        } else {
          lock.unlock();
          uint8_t ok = 1;
          writeAll(fd, &ok, 1);
        }
      } else {
        break;
      }
    }
    close(fd);
  }

  int listenFd_;
  std::atomic<bool> done_{false};
  std::thread acceptThread_;
  std::mutex mu_;
  std::condition_variable cv_;
  std::map<std::string, std::vector<char>> map_;
  std::vector<int> clientFds_;
  std::vector<std::thread> clientThreads_;
};

TcpStore::TcpStore(
    const std::string& host,
    int port,
    bool isServer,
    std::chrono::milliseconds timeout)
    : timeout_(timeout) {
  if (isServer) {
    server_ = std::make_unique<Server>(port);
  }
  // Connect as client (with retries while the server comes up elsewhere).
  struct addrinfo hints;
  std::memset(&hints, 0, sizeof(hints));
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  struct addrinfo* res = nullptr;
  int rv = getaddrinfo(host.c_str(), std::to_string(port).c_str(), &hints, &res);
  GA_ENFORCE_EQ(rv, 0, "getaddrinfo(", host, "): ", gai_strerror(rv));
  auto deadline = std::chrono::steady_clock::now() + timeout;
  for (;;) {
    clientFd_ = socket(res->ai_family, SOCK_STREAM | SOCK_CLOEXEC, 0);
    GA_ENFORCE_GE(clientFd_, 0);
    if (::connect(clientFd_, res->ai_addr, res->ai_addrlen) == 0) {
      break;
    }
    close(clientFd_);
    clientFd_ = -1;
    if (std::chrono::steady_clock::now() > deadline) {
      freeaddrinfo(res);
      throw TimeoutException(
          "TcpStore: cannot connect to " + host + ":" + std::to_string(port));
    }
    usleep(50 * 1000);
  }
  freeaddrinfo(res);
  int on = 1;
  setsockopt(clientFd_, IPPROTO_TCP, TCP_NODELAY, &on, sizeof(on));
}

TcpStore::~TcpStore() {
  if (clientFd_ >= 0) {
    close(clientFd_);
  }
}

void TcpStore::set(const std::string& key, const std::vector<char>& data) {
  std::lock_guard<std::mutex> lock(clientMu_);
  uint8_t op = 'S';
  writeAll(clientFd_, &op, 1);
  writeString(clientFd_, key);
  uint32_t vlen = data.size();
  writeAll(clientFd_, &vlen, 4);
  if (vlen > 0) {
    writeAll(clientFd_, data.data(), vlen);
  }
}

void TcpStore::append(
    const std::string& key,
    const std::vector<char>& data) {
  std::lock_guard<std::mutex> lock(clientMu_);
  uint8_t op = 'A';
  writeAll(clientFd_, &op, 1);
  writeString(clientFd_, key);
  uint32_t vlen = data.size();
  writeAll(clientFd_, &vlen, 4);
  if (vlen > 0) {
    writeAll(clientFd_, data.data(), vlen);
  }
}

int64_t TcpStore::add(const std::string& key, int64_t delta) {
  std::lock_guard<std::mutex> lock(clientMu_);
  uint8_t op = 'I';
  writeAll(clientFd_, &op, 1);
  writeString(clientFd_, key);
  writeAll(clientFd_, &delta, 8);
  int64_t result;
  GA_ENFORCE(readAll(clientFd_, &result, 8), "TcpStore: server closed");
  return result;
}

std::vector<char> TcpStore::get(const std::string& key) {
  // The 'G' opcode blocks server-side on a condvar with no deadline;
  // bound it with the store timeout by first polling for existence
  // (wait honors deadlines client-side), then fetching.
  wait({key}, timeout_);
  std::lock_guard<std::mutex> lock(clientMu_);
  uint8_t op = 'G';
  writeAll(clientFd_, &op, 1);
  writeString(clientFd_, key);
  uint32_t vlen;
  GA_ENFORCE(readAll(clientFd_, &vlen, 4), "TcpStore: server closed");
  std::vector<char> val(vlen);
  if (vlen > 0) {
    GA_ENFORCE(readAll(clientFd_, val.data(), vlen), "TcpStore: server closed");
  }
  return val;
}

void TcpStore::wait(
    const std::vector<std::string>& keys,
    const std::chrono::milliseconds& timeout) {
  // Poll the non-blocking check op so the client honors its deadline
  // (a blocked server-side 'W' would leave a stray reply in the stream
  // if the client gave up). Negative timeout = wait forever.
  auto deadline = std::chrono::steady_clock::now() + timeout;
  for (const auto& key : keys) {
    for (;;) {
      uint8_t exists;
      {
        std::lock_guard<std::mutex> lock(clientMu_);
        uint8_t op = 'C';
        writeAll(clientFd_, &op, 1);
        writeString(clientFd_, key);
        GA_ENFORCE(
            readAll(clientFd_, &exists, 1), "TcpStore: server closed");
      }
      if (exists) {
        break;
      }
      if (timeout.count() >= 0 &&
          std::chrono::steady_clock::now() > deadline) {
        throw TimeoutException("TcpStore::wait timeout for key " + key);
      }
      usleep(10 * 1000);
    }
  }
}

} // namespace glooamd
