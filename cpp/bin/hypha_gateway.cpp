// hypha-gateway: registry + pub/sub broker + health (crates/gateway role).

#include <signal.h>

#include <chrono>
#include <string>
#include <thread>

#include "hypha/gateway.h"

int main(int argc, char** argv) {
  int port = 41000;
  hypha::TlsConfig tls;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    if (a == "--port") port = std::stoi(argv[++i]);
    else if (a == "--tls-cert") tls.cert_path = argv[++i];
    else if (a == "--tls-key") tls.key_path = argv[++i];
    else if (a == "--tls-ca") tls.ca_path = argv[++i];
    else if (a == "--tls-crl") tls.crl_path = argv[++i];
  }
  signal(SIGPIPE, SIG_IGN);
  hypha::Gateway gw(tls);
  gw.start(port);
  printf("hypha-gateway on port %d\n", gw.port());
  fflush(stdout);
  while (true) std::this_thread::sleep_for(std::chrono::seconds(3600));
}
