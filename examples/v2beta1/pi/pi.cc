// pi — Monte-Carlo π estimation, the CPU smoke workload of the stack
// (BASELINE.json config 1; fills the role of the reference's
// examples/v2beta1/pi/pi.cc, which is an MPI program).
//
// Unlike the reference this is NOT an MPI program: the MI355X-native stack's
// boot plane is amdrun (ssh + rendezvous env), so ranks learn their identity
// from RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT and reduce their hit counts to
// rank 0 over one TCP connection each — the same env contract every workload
// of the stack (including the RCCL trainers) boots from.
//
// Build:  g++ -O2 -o pi pi.cc
// Run:    amdrun -np 2 --hostfile /etc/mpi/hostfile -- /opt/pi/pi

#include <arpa/inet.h>
#include <netdb.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cerrno>
#include <cinttypes>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <random>

static int env_int(const char* name, int dflt) {
  const char* v = getenv(name);
  return v ? atoi(v) : dflt;
}

static const char* env_str(const char* name, const char* dflt) {
  const char* v = getenv(name);
  return v ? v : dflt;
}

// Rank 0 listens on MASTER_PORT+1 and sums one int64 from every peer;
// peers connect (with retry — workers may start at different times) and
// send theirs. This is the stack's minimal "MPI_Reduce(SUM, root 0)".
static int64_t reduce_to_root(int rank, int world, int64_t mine) {
  int port = env_int("MASTER_PORT", 29500) + 1;
  if (world == 1) return mine;
  if (rank == 0) {
    int srv = socket(AF_INET, SOCK_STREAM, 0);
    int one = 1;
    setsockopt(srv, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = INADDR_ANY;
    addr.sin_port = htons((uint16_t)port);
    if (bind(srv, (sockaddr*)&addr, sizeof(addr)) != 0 || listen(srv, world) != 0) {
      fprintf(stderr, "pi: rank 0 cannot listen on %d: %s\n", port, strerror(errno));
      exit(1);
    }
    int64_t total = mine;
    for (int i = 1; i < world; i++) {
      int c = accept(srv, nullptr, nullptr);
      int64_t v = 0;
      size_t got = 0;
      while (got < sizeof(v)) {
        ssize_t n = read(c, (char*)&v + got, sizeof(v) - got);
        if (n <= 0) { fprintf(stderr, "pi: short read from peer\n"); exit(1); }
        got += (size_t)n;
      }
      total += v;
      close(c);
    }
    close(srv);
    return total;
  }
  const char* master = env_str("MASTER_ADDR", "127.0.0.1");
  addrinfo hints{}, *res = nullptr;
  hints.ai_family = AF_INET;
  hints.ai_socktype = SOCK_STREAM;
  char portstr[16];
  snprintf(portstr, sizeof(portstr), "%d", port);
  for (int attempt = 0; attempt < 60; attempt++) {
    if (res == nullptr && getaddrinfo(master, portstr, &hints, &res) != 0) {
      sleep(1);
      continue;
    }
    int fd = socket(AF_INET, SOCK_STREAM, 0);
    if (connect(fd, res->ai_addr, res->ai_addrlen) == 0) {
      size_t sent = 0;
      while (sent < sizeof(mine)) {
        ssize_t n = write(fd, (char*)&mine + sent, sizeof(mine) - sent);
        if (n <= 0) break;
        sent += (size_t)n;
      }
      close(fd);
      freeaddrinfo(res);
      return 0;
    }
    close(fd);
    sleep(1);  // rank 0 not up yet — retry, like ssh ConnectionAttempts
  }
  fprintf(stderr, "pi: cannot reach rank 0 at %s:%d\n", master, port);
  exit(1);
}

int main(int argc, char** argv) {
  int rank = env_int("RANK", env_int("OMPI_COMM_WORLD_RANK", 0));
  int world = env_int("WORLD_SIZE", env_int("OMPI_COMM_WORLD_SIZE", 1));
  int64_t samples = argc > 1 ? atoll(argv[1]) : 10000000LL;

  char host[256];
  gethostname(host, sizeof(host));
  printf("pi: rank %d/%d on %s\n", rank, world, host);
  fflush(stdout);

  std::mt19937_64 gen(0x5deece66dULL + (uint64_t)rank);
  std::uniform_real_distribution<double> uni(0.0, 1.0);
  int64_t inside = 0;
  for (int64_t i = 0; i < samples; i++) {
    double x = uni(gen), y = uni(gen);
    if (x * x + y * y <= 1.0) inside++;
  }

  int64_t total = reduce_to_root(rank, world, inside);
  if (rank == 0) {
    double pi = 4.0 * (double)total / ((double)samples * world);
    printf("pi is approximately %.8f (%" PRId64 " samples over %d ranks)\n",
           pi, samples * world, world);
  }
  return 0;
}
