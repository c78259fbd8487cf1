// modelxd — the registry HTTP server (C++ native).
// Same flag surface as the reference (cmd/modelxd/modelxd.go:44-55):
//   --listen --s3-url --s3-bucket --s3-access-key --s3-secret-key --s3-region
//   --s3-presign-expire --enable-redirect --local-data
// plus offline auth: --auth-tokens, --jwt-hs256-secret.
#include <signal.h>
#include <unistd.h>

#include <atomic>
#include <chrono>
#include <thread>
#include <cstdio>
#include <cstring>
#include <memory>

#include "modelx/http.hpp"
#include "modelx/registry.hpp"
#include "modelx/s3.hpp"

using namespace modelx;

int main(int argc, char** argv) {
  std::string listen = ":8080";
  std::string local_data = "data/registry";
  store::S3Options s3;
  bool use_s3 = false;
  bool enable_redirect = false;
  registry::AuthConfig auth;
  http::TlsConfig tls;
  int gc_interval_s = 0;

  for (int i = 1; i < argc; i++) {
    std::string a = argv[i];
    auto next = [&]() -> std::string { return i + 1 < argc ? argv[++i] : ""; };
    if (a == "--listen") listen = next();
    else if (a == "--local-data") local_data = next();
    else if (a == "--s3-url") { s3.endpoint = next(); use_s3 = true; }
    else if (a == "--s3-public-url") s3.public_endpoint = next();
    else if (a == "--s3-bucket") s3.bucket = next();
    else if (a == "--s3-access-key") s3.access_key = next();
    else if (a == "--s3-secret-key") s3.secret_key = next();
    else if (a == "--s3-region") s3.region = next();
    else if (a == "--s3-presign-expire") s3.presign_expire_seconds = atoi(next().c_str());
    else if (a == "--enable-redirect") enable_redirect = true;
    else if (a == "--auth-tokens") {
      std::string toks = next();
      size_t pos = 0;
      while (pos <= toks.size()) {
        size_t comma = toks.find(',', pos);
        if (comma == std::string::npos) comma = toks.size();
        if (comma > pos) auth.tokens.push_back(toks.substr(pos, comma - pos));
        pos = comma + 1;
      }
    } else if (a == "--jwt-hs256-secret") auth.jwt_hs256_secret = next();
    else if (a == "--oidc-jwks") {
      std::string path = next();
      std::string err;
      if (!registry::load_jwks_file(path, &auth.jwks, &err)) {
        fprintf(stderr, "modelxd: --oidc-jwks: %s\n", err.c_str());
        return 1;
      }
    } else if (a == "--oidc-issuer") auth.oidc_issuer = next();
    else if (a == "--oidc-audience") auth.oidc_audience = next();
    else if (a == "--gc-interval") gc_interval_s = atoi(next().c_str());
    else if (a == "--tls-cert") tls.cert_file = next();
    else if (a == "--tls-key") tls.key_file = next();
    else if (a == "--help" || a == "-h") {
      printf("modelxd: modelx registry server (MI355X-native build)\n"
             "  --listen :8080            listen address\n"
             "  --local-data DIR          local FS backend root (default data/registry)\n"
             "  --s3-url URL              S3 endpoint (switches to S3 backend)\n"
             "  --s3-public-url URL       endpoint presigned URLs point at (default --s3-url)\n"
             "  --s3-bucket B --s3-access-key K --s3-secret-key S --s3-region R\n"
             "  --s3-presign-expire SECS  presigned URL lifetime (default 3600)\n"
             "  --enable-redirect         hand out presigned S3 locations\n"
             "  --auth-tokens T1,T2       static bearer tokens\n"
             "  --jwt-hs256-secret S      verify HS256 JWTs offline\n"
             "  --oidc-jwks FILE          verify RS256 ID tokens against a JWKS document\n"
             "  --oidc-issuer ISS         require `iss` claim to equal ISS\n"
             "  --oidc-audience AUD       require `aud` claim to contain AUD\n"
             "  --gc-interval SECS        periodic mark-sweep GC of all repositories\n"
             "                            (reference has manual POST garbage-collect only)\n"
             "  --tls-cert F --tls-key F  serve HTTPS (reference --tls-*, server.go:37-43)\n");
      return 0;
    }
  }
  signal(SIGPIPE, SIG_IGN);

  std::shared_ptr<store::RegistryStore> st;
  if (use_s3) {
    auto provider = std::make_shared<store::S3FSProvider>(s3);
    if (enable_redirect) {
      st = std::make_shared<store::S3RegistryStore>(provider);
    } else {
      st = std::make_shared<store::RegistryStore>(provider);
    }
    // initial global index rebuild (server.go:46-63 → RefreshGlobalIndex)
    st->RefreshGlobalIndex();
  } else {
    auto provider = std::make_shared<store::LocalFSProvider>(local_data);
    st = std::make_shared<store::RegistryStore>(provider);
    st->RefreshGlobalIndex();
  }
  registry::Registry reg(st, auth);

  http::Server server(listen, [&reg](http::Request& req, http::ResponseWriter& w) {
    auto t0 = std::chrono::steady_clock::now();
    reg.handle(req, w);
    auto us = std::chrono::duration_cast<std::chrono::microseconds>(
                  std::chrono::steady_clock::now() - t0)
                  .count();
    // request logging filter (reference: pkg/registry/helper.go:98-113)
    fprintf(stderr, "%s %s %lldus %s\n", req.method.c_str(), req.target.c_str(),
            static_cast<long long>(us), req.client_addr.c_str());
  }, tls);
  int port = server.start();
  printf("modelxd listening on port %d backend=%s redirect=%d\n", port,
         use_s3 ? "s3" : "local", enable_redirect ? 1 : 0);
  fflush(stdout);
  // scheduled GC sweep (reference ships manual POST garbage-collect only;
  // operational gap noted in docs/roadmap.md). Runs the same mark-sweep the
  // endpoint runs; content-addressed writes make sweeps safe to repeat.
  std::atomic<bool> gc_stop{false};
  std::thread gc_thread;
  if (gc_interval_s > 0) {
    gc_thread = std::thread([&] {
      while (!gc_stop.load()) {
        for (int i = 0; i < gc_interval_s * 10 && !gc_stop.load(); i++)
          std::this_thread::sleep_for(std::chrono::milliseconds(100));
        if (gc_stop.load()) break;
        int removed = st->GCBlobsAll();
        fprintf(stderr, "gc sweep: %d blob(s) removed\n", removed);
      }
    });
  }
  sigset_t set;
  sigemptyset(&set);
  sigaddset(&set, SIGINT);
  sigaddset(&set, SIGTERM);
  sigprocmask(SIG_BLOCK, &set, nullptr);
  int sig = 0;
  sigwait(&set, &sig);
  gc_stop.store(true);
  if (gc_thread.joinable()) gc_thread.join();
  server.stop();  // graceful shutdown (server.go:33-36)
  return 0;
}
