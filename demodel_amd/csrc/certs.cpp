// demodel_amd._native — native CA / leaf-certificate minting via libcrypto.
//
// Rebuilds the reference's trust layer natively (reference: Go crypto/x509 in
// cmd/demodel/init.go:64-148 for the CA, cmd/demodel/start.go:49-122 for
// per-host leaf minting).  Semantics kept:
//   * CA: self-signed, CA:true pathlen:0, keyUsage certSign|cRLSign,
//     SKID = SHA-1(SPKI), validity 2 years + 3 months (init.go:92-115).
//   * Leaf: EKU serverAuth+clientAuth, SAN DNS:<host> (or IP:<host>),
//     validity 2y3m, signed by the CA (start.go:69-94).
// Fixed vs reference: RSA keys are 4096 bits (the reference generates 4095 —
// init.go:69 quirk, flagged in SURVEY.md §2.1).
//
// Build: c++ -shared -fPIC -O2 certs.cpp -lcrypto (see demodel_amd/build.py).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <openssl/bio.h>
#include <openssl/bn.h>
#include <openssl/err.h>
#include <openssl/evp.h>
#include <openssl/pem.h>
#include <openssl/rand.h>
#include <openssl/x509.h>
#include <openssl/x509v3.h>

#include <memory>
#include <stdexcept>
#include <string>

namespace py = pybind11;

namespace {

[[noreturn]] void throw_ssl(const std::string& what) {
  char buf[256];
  ERR_error_string_n(ERR_get_error(), buf, sizeof(buf));
  throw std::runtime_error(what + ": " + buf);
}

struct EvpKeyDel { void operator()(EVP_PKEY* p) const { EVP_PKEY_free(p); } };
struct X509Del   { void operator()(X509* p) const { X509_free(p); } };
struct BioDel    { void operator()(BIO* p) const { BIO_free(p); } };
using KeyPtr  = std::unique_ptr<EVP_PKEY, EvpKeyDel>;
using CertPtr = std::unique_ptr<X509, X509Del>;
using BioPtr  = std::unique_ptr<BIO, BioDel>;

KeyPtr gen_key(bool ecdsa) {
  EVP_PKEY* k = ecdsa ? EVP_EC_gen("P-256") : EVP_RSA_gen(4096);
  if (!k) throw_ssl("key generation failed");
  return KeyPtr(k);
}

// 128-bit random serial (reference main.go:49-54 randomSerialNumber).
void set_random_serial(X509* crt) {
  unsigned char raw[16];
  if (RAND_bytes(raw, sizeof(raw)) != 1) throw_ssl("RAND_bytes");
  raw[0] &= 0x7f;  // keep it positive
  BIGNUM* bn = BN_bin2bn(raw, sizeof(raw), nullptr);
  ASN1_INTEGER* serial = X509_get_serialNumber(crt);
  BN_to_ASN1_INTEGER(bn, serial);
  BN_free(bn);
}

// 2 years + 3 months, like the reference (init.go:101, start.go:80).
constexpr long kValiditySecs = (2L * 365 + 90) * 24 * 3600;

void add_ext(X509* crt, X509* issuer, int nid, const char* value) {
  X509V3_CTX ctx;
  X509V3_set_ctx_nodb(&ctx);
  X509V3_set_ctx(&ctx, issuer, crt, nullptr, nullptr, 0);
  X509_EXTENSION* ex = X509V3_EXT_conf_nid(nullptr, &ctx, nid, value);
  if (!ex) throw_ssl(std::string("extension ") + value);
  X509_add_ext(crt, ex, -1);
  X509_EXTENSION_free(ex);
}

std::string pem_cert(X509* crt) {
  BioPtr bio(BIO_new(BIO_s_mem()));
  if (PEM_write_bio_X509(bio.get(), crt) != 1) throw_ssl("PEM cert");
  char* data;
  long n = BIO_get_mem_data(bio.get(), &data);
  return std::string(data, n);
}

std::string pem_key(EVP_PKEY* key) {
  BioPtr bio(BIO_new(BIO_s_mem()));
  if (PEM_write_bio_PrivateKey(bio.get(), key, nullptr, nullptr, 0, nullptr,
                               nullptr) != 1)
    throw_ssl("PEM key");
  char* data;
  long n = BIO_get_mem_data(bio.get(), &data);
  return std::string(data, n);
}

CertPtr parse_cert(const std::string& pem) {
  BioPtr bio(BIO_new_mem_buf(pem.data(), (int)pem.size()));
  X509* crt = PEM_read_bio_X509(bio.get(), nullptr, nullptr, nullptr);
  if (!crt) throw_ssl("parse CA cert");
  return CertPtr(crt);
}

KeyPtr parse_key(const std::string& pem) {
  BioPtr bio(BIO_new_mem_buf(pem.data(), (int)pem.size()));
  EVP_PKEY* k = PEM_read_bio_PrivateKey(bio.get(), nullptr, nullptr, nullptr);
  if (!k) throw_ssl("parse CA key");
  return KeyPtr(k);
}

void set_name(X509_NAME* name, const char* cn, const char* org) {
  X509_NAME_add_entry_by_txt(name, "O", MBSTRING_ASC,
                             (const unsigned char*)org, -1, -1, 0);
  X509_NAME_add_entry_by_txt(name, "CN", MBSTRING_ASC,
                             (const unsigned char*)cn, -1, -1, 0);
}

bool looks_like_ip(const std::string& h) {
  bool digit_dot = !h.empty();
  for (char c : h)
    if (!isdigit((unsigned char)c) && c != '.') { digit_dot = false; break; }
  return digit_dot || h.find(':') != std::string::npos;
}

std::pair<std::string, std::string> ca_create(bool ecdsa) {
  KeyPtr key = gen_key(ecdsa);
  CertPtr crt(X509_new());
  X509_set_version(crt.get(), 2);
  set_random_serial(crt.get());
  X509_gmtime_adj(X509_getm_notBefore(crt.get()), 0);
  X509_gmtime_adj(X509_getm_notAfter(crt.get()), kValiditySecs);
  set_name(X509_get_subject_name(crt.get()), "demodel-amd Root CA",
           "demodel-amd");
  X509_set_issuer_name(crt.get(), X509_get_subject_name(crt.get()));
  X509_set_pubkey(crt.get(), key.get());
  add_ext(crt.get(), crt.get(), NID_basic_constraints,
          "critical,CA:TRUE,pathlen:0");
  add_ext(crt.get(), crt.get(), NID_key_usage, "critical,keyCertSign,cRLSign");
  add_ext(crt.get(), crt.get(), NID_subject_key_identifier, "hash");
  if (X509_sign(crt.get(), key.get(), EVP_sha256()) == 0) throw_ssl("CA sign");
  return {pem_cert(crt.get()), pem_key(key.get())};
}

std::pair<std::string, std::string> leaf_create(const std::string& ca_cert_pem,
                                                const std::string& ca_key_pem,
                                                const std::string& hostname,
                                                bool ecdsa) {
  CertPtr ca_crt = parse_cert(ca_cert_pem);
  KeyPtr ca_key = parse_key(ca_key_pem);
  KeyPtr key = gen_key(ecdsa);

  CertPtr crt(X509_new());
  X509_set_version(crt.get(), 2);
  set_random_serial(crt.get());
  X509_gmtime_adj(X509_getm_notBefore(crt.get()), -300);  // clock-skew slack
  X509_gmtime_adj(X509_getm_notAfter(crt.get()), kValiditySecs);
  set_name(X509_get_subject_name(crt.get()), hostname.c_str(), "demodel-amd");
  X509_set_issuer_name(crt.get(), X509_get_subject_name(ca_crt.get()));
  X509_set_pubkey(crt.get(), key.get());
  add_ext(crt.get(), ca_crt.get(), NID_basic_constraints, "critical,CA:FALSE");
  add_ext(crt.get(), ca_crt.get(), NID_key_usage,
          "critical,digitalSignature,keyEncipherment");
  add_ext(crt.get(), ca_crt.get(), NID_ext_key_usage,
          "serverAuth,clientAuth");
  std::string san = (looks_like_ip(hostname) ? "IP:" : "DNS:") + hostname;
  add_ext(crt.get(), ca_crt.get(), NID_subject_alt_name, san.c_str());
  add_ext(crt.get(), ca_crt.get(), NID_authority_key_identifier,
          "keyid:always");
  if (X509_sign(crt.get(), ca_key.get(), EVP_sha256()) == 0)
    throw_ssl("leaf sign");
  return {pem_cert(crt.get()), pem_key(key.get())};
}

}  // namespace

void register_zstd_host(py::module_& m);

PYBIND11_MODULE(_native, m) {
  m.doc() = "demodel-amd native helpers (libcrypto cert minting, "
            "host zstd reference decoder)";
  register_zstd_host(m);
  m.def("ca_create", &ca_create, py::arg("ecdsa") = false,
        py::call_guard<py::gil_scoped_release>(),
        "Generate a self-signed demodel CA -> (cert_pem, key_pem)");
  m.def("leaf_create", &leaf_create, py::arg("ca_cert_pem"),
        py::arg("ca_key_pem"), py::arg("hostname"), py::arg("ecdsa") = false,
        py::call_guard<py::gil_scoped_release>(),
        "Mint a per-host leaf certificate signed by the CA -> "
        "(cert_pem, key_pem)");
}
