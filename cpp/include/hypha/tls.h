// Mutual-TLS context for the control-plane transport.
// Parity with the reference's security model (rfc/2025-05-30_mtls.md,
// docs/security.md): every connection is mutually authenticated against the
// deployment's CA chain (tools/hypha_certutil.py generates the 3-tier
// Ed25519 PKI); peer identity is the certificate, not the `from` field.
#pragma once

#include <openssl/err.h>
#include <openssl/pem.h>
#include <openssl/ssl.h>
#include <openssl/x509.h>

#include <memory>
#include <stdexcept>
#include <string>

namespace hypha {

struct TlsConfig {
  std::string cert_path;  // node certificate (PEM; may be a chain bundle)
  std::string key_path;   // node private key (PEM)
  std::string ca_path;    // trust anchor bundle (root + org CAs)
  std::string crl_path;   // optional CRL bundle (PEM; one CRL per issuing CA)
  bool enabled() const { return !cert_path.empty(); }
};

class TlsContext {
 public:
  explicit TlsContext(const TlsConfig& cfg) {
    ctx_ = SSL_CTX_new(TLS_method());
    if (!ctx_) throw std::runtime_error("tls: SSL_CTX_new failed");
    SSL_CTX_set_min_proto_version(ctx_, TLS1_3_VERSION);
    if (SSL_CTX_use_certificate_chain_file(ctx_, cfg.cert_path.c_str()) != 1)
      throw std::runtime_error("tls: cannot load cert " + cfg.cert_path);
    if (SSL_CTX_use_PrivateKey_file(ctx_, cfg.key_path.c_str(), SSL_FILETYPE_PEM) != 1)
      throw std::runtime_error("tls: cannot load key " + cfg.key_path);
    if (SSL_CTX_check_private_key(ctx_) != 1)
      throw std::runtime_error("tls: key does not match cert");
    if (SSL_CTX_load_verify_locations(ctx_, cfg.ca_path.c_str(), nullptr) != 1)
      throw std::runtime_error("tls: cannot load CA " + cfg.ca_path);
    // mutual authentication: both sides must present a CA-signed cert
    SSL_CTX_set_verify(ctx_, SSL_VERIFY_PEER | SSL_VERIFY_FAIL_IF_NO_PEER_CERT,
                       nullptr);
    // Revocation: the reference's forked libp2p checks CRLs via rustls
    // WebPkiClientVerifier (rfc/2025-05-30_mtls.md). When a CRL bundle is
    // configured, every peer leaf cert is checked against its issuer's CRL
    // (X509_V_FLAG_CRL_CHECK: leaf-only, so only issuing org CAs need CRLs).
    if (!cfg.crl_path.empty()) {
      X509_STORE* store = SSL_CTX_get_cert_store(ctx_);
      FILE* fp = fopen(cfg.crl_path.c_str(), "r");
      if (!fp) throw std::runtime_error("tls: cannot open CRL " + cfg.crl_path);
      int loaded = 0;
      while (X509_CRL* crl = PEM_read_X509_CRL(fp, nullptr, nullptr, nullptr)) {
        X509_STORE_add_crl(store, crl);
        X509_CRL_free(crl);
        ++loaded;
      }
      fclose(fp);
      if (loaded == 0)
        throw std::runtime_error("tls: no CRLs found in " + cfg.crl_path);
      X509_STORE_set_flags(store, X509_V_FLAG_CRL_CHECK);
    }
    // No TLS1.3 session tickets: a client that only WRITES (tensor push
    // streams) would otherwise leave the server's post-handshake ticket
    // records unread, and closing a socket with unread data sends TCP RST —
    // destroying the in-flight payload.
    SSL_CTX_set_num_tickets(ctx_, 0);
    SSL_CTX_set_options(ctx_, SSL_OP_NO_TICKET);
  }
  ~TlsContext() {
    if (ctx_) SSL_CTX_free(ctx_);
  }
  TlsContext(const TlsContext&) = delete;

  // returns nullptr on handshake failure (connection must be closed)
  SSL* wrap(int fd, bool server) const {
    SSL* ssl = SSL_new(ctx_);
    if (!ssl) return nullptr;
    SSL_set_fd(ssl, fd);
    int rc = server ? SSL_accept(ssl) : SSL_connect(ssl);
    if (rc != 1) {
      SSL_free(ssl);
      return nullptr;
    }
    return ssl;
  }

  // peer identity: CN of the verified peer certificate ("" if none)
  static std::string peer_cn(SSL* ssl) {
    if (!ssl) return "";
    X509* cert = SSL_get_peer_certificate(ssl);
    if (!cert) return "";
    char buf[256] = {0};
    X509_NAME_get_text_by_NID(X509_get_subject_name(cert), NID_commonName, buf,
                              sizeof buf - 1);
    X509_free(cert);
    return buf;
  }

 private:
  SSL_CTX* ctx_ = nullptr;
};

}  // namespace hypha
