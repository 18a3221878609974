// hypha-gateway: registry + pub/sub broker + health (crates/gateway role).

#include <signal.h>

#include <chrono>
#include <string>
#include <thread>

#include "hypha/gateway.h"

int main(int argc, char** argv) {
  int port = 41000;
  std::string listen_host = "127.0.0.1";
  bool probe = false, init = false;
  hypha::TlsConfig tls;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    if (a == "--port") port = std::stoi(argv[++i]);
    else if (a == "--listen-host") listen_host = argv[++i];
    else if (a == "--tls-cert") tls.cert_path = argv[++i];
    else if (a == "--tls-key") tls.key_path = argv[++i];
    else if (a == "--tls-ca") tls.ca_path = argv[++i];
    else if (a == "--tls-crl") tls.crl_path = argv[++i];
    else if (a == "probe") probe = true;
    else if (a == "init") init = true;
  }
  signal(SIGPIPE, SIG_IGN);
  if (init) {  // reference CLI Init subcommand: emit a commented config
    printf("# hypha-gateway configuration (flags)\n"
           "# --port 41000        broker listen port\n"
           "# --listen-host 0.0.0.0  bind address (default loopback-only;\n"
           "#                     0.0.0.0 accepts off-host peers)\n"
           "# --tls-cert/--tls-key/--tls-ca [--tls-crl]  mTLS identity\n");
    return 0;
  }
  if (probe) {  // readiness: is a gateway listening on the port?
    int fd = hypha::tcp_connect("127.0.0.1", port, 2.0);
    if (fd < 0) {
      fprintf(stderr, "probe: gateway unreachable on port %d\n", port);
      return 1;
    }
    close(fd);
    printf("probe: healthy\n");
    return 0;
  }
  hypha::Gateway gw(tls);
  gw.start(port, listen_host);
  printf("hypha-gateway on port %d\n", gw.port());
  fflush(stdout);
  while (true) std::this_thread::sleep_for(std::chrono::seconds(3600));
}
