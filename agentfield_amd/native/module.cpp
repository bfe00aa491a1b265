// agentfield_amd._native — CPU-side native runtime components:
//   * Ed25519 keygen/sign/verify via OpenSSL libcrypto (DID/VC audit path;
//     reference does this in Go crypto/ed25519 — SURVEY.md C21/C22)
//   * the continuous-batching scheduler + KV page allocator (the production
//     counterpart of engine/scheduler.py, which stays as the tested oracle)
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <openssl/evp.h>
#include <openssl/rand.h>

#include <deque>
#include <list>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace py = pybind11;

// ------------------------------- ed25519 ----------------------------------
static py::bytes ed25519_pubkey(py::bytes seed) {
  std::string s = seed;
  if (s.size() != 32) throw std::invalid_argument("seed must be 32 bytes");
  EVP_PKEY* k = EVP_PKEY_new_raw_private_key(EVP_PKEY_ED25519, nullptr,
                                             (const unsigned char*)s.data(), 32);
  if (!k) throw std::runtime_error("ed25519 key import failed");
  unsigned char pub[32];
  size_t plen = 32;
  EVP_PKEY_get_raw_public_key(k, pub, &plen);
  EVP_PKEY_free(k);
  return py::bytes((const char*)pub, plen);
}

static py::bytes ed25519_sign(py::bytes seed, py::bytes msg) {
  std::string s = seed, m = msg;
  if (s.size() != 32) throw std::invalid_argument("seed must be 32 bytes");
  EVP_PKEY* k = EVP_PKEY_new_raw_private_key(EVP_PKEY_ED25519, nullptr,
                                             (const unsigned char*)s.data(), 32);
  if (!k) throw std::runtime_error("ed25519 key import failed");
  EVP_MD_CTX* ctx = EVP_MD_CTX_new();
  unsigned char sig[64];
  size_t slen = 64;
  int ok = EVP_DigestSignInit(ctx, nullptr, nullptr, nullptr, k) == 1 &&
           EVP_DigestSign(ctx, sig, &slen, (const unsigned char*)m.data(),
                          m.size()) == 1;
  EVP_MD_CTX_free(ctx);
  EVP_PKEY_free(k);
  if (!ok) throw std::runtime_error("ed25519 sign failed");
  return py::bytes((const char*)sig, slen);
}

static bool ed25519_verify(py::bytes pub, py::bytes msg, py::bytes sig) {
  std::string p = pub, m = msg, g = sig;
  if (p.size() != 32 || g.size() != 64) return false;
  EVP_PKEY* k = EVP_PKEY_new_raw_public_key(EVP_PKEY_ED25519, nullptr,
                                            (const unsigned char*)p.data(), 32);
  if (!k) return false;
  EVP_MD_CTX* ctx = EVP_MD_CTX_new();
  bool ok = EVP_DigestVerifyInit(ctx, nullptr, nullptr, nullptr, k) == 1 &&
            EVP_DigestVerify(ctx, (const unsigned char*)g.data(), g.size(),
                             (const unsigned char*)m.data(), m.size()) == 1;
  EVP_MD_CTX_free(ctx);
  EVP_PKEY_free(k);
  return ok;
}

// ------------------------------- AES-256-GCM -------------------------------
static py::bytes aes_gcm_encrypt(py::bytes key, py::bytes plaintext) {
  std::string k = key, pt = plaintext;
  if (k.size() != 32) throw std::invalid_argument("key must be 32 bytes");
  unsigned char iv[12];
  if (RAND_bytes(iv, 12) != 1) throw std::runtime_error("RAND_bytes failed");
  std::string out(12 + pt.size() + 16, '\0');
  memcpy(&out[0], iv, 12);
  EVP_CIPHER_CTX* ctx = EVP_CIPHER_CTX_new();
  int len = 0, total = 0;
  bool ok = EVP_EncryptInit_ex(ctx, EVP_aes_256_gcm(), nullptr,
                               (const unsigned char*)k.data(), iv) == 1 &&
            EVP_EncryptUpdate(ctx, (unsigned char*)&out[12], &len,
                              (const unsigned char*)pt.data(),
                              (int)pt.size()) == 1;
  total = len;
  ok = ok && EVP_EncryptFinal_ex(ctx, (unsigned char*)&out[12 + total], &len) == 1;
  total += len;
  unsigned char tag[16];
  ok = ok && EVP_CIPHER_CTX_ctrl(ctx, EVP_CTRL_GCM_GET_TAG, 16, tag) == 1;
  EVP_CIPHER_CTX_free(ctx);
  if (!ok) throw std::runtime_error("aes-gcm encrypt failed");
  memcpy(&out[12 + total], tag, 16);
  out.resize(12 + total + 16);
  return py::bytes(out);
}

static py::bytes aes_gcm_decrypt(py::bytes key, py::bytes blob) {
  std::string k = key, b = blob;
  if (k.size() != 32) throw std::invalid_argument("key must be 32 bytes");
  if (b.size() < 28) throw std::invalid_argument("ciphertext too short");
  const unsigned char* iv = (const unsigned char*)b.data();
  const unsigned char* ct = iv + 12;
  size_t ctlen = b.size() - 12 - 16;
  const unsigned char* tag = (const unsigned char*)b.data() + b.size() - 16;
  std::string out(ctlen, '\0');
  EVP_CIPHER_CTX* ctx = EVP_CIPHER_CTX_new();
  int len = 0;
  bool ok = EVP_DecryptInit_ex(ctx, EVP_aes_256_gcm(), nullptr,
                               (const unsigned char*)k.data(), iv) == 1 &&
            EVP_DecryptUpdate(ctx, (unsigned char*)&out[0], &len, ct,
                              (int)ctlen) == 1 &&
            EVP_CIPHER_CTX_ctrl(ctx, EVP_CTRL_GCM_SET_TAG, 16,
                                (void*)tag) == 1;
  int fin = 0;
  ok = ok && EVP_DecryptFinal_ex(ctx, (unsigned char*)&out[0] + len, &fin) == 1;
  EVP_CIPHER_CTX_free(ctx);
  if (!ok) throw std::runtime_error("aes-gcm decrypt failed (bad key or tampered)");
  out.resize(len + fin);
  return py::bytes(out);
}

// ------------------------- scheduler / allocator ---------------------------
// Semantics mirror engine/scheduler.py exactly (it is the tested oracle).
struct SeqState {
  int prompt_len = 0;
  int num_tokens = 0;   // prompt + generated
  int cached_prefix = 0;  // prompt tokens served from the prefix cache
  int freed_pages = 0;  // leading pages reclaimed by the rolling window
  std::vector<int> pages;
  std::vector<long> hashes;  // chained per-full-page prompt hashes
};

struct ScheduleResult {
  bool has_work = false;
  bool is_prefill = false;
  std::vector<long> seq_ids;
  std::vector<long> preempted;  // seqs to re-prefill (outputs retained)
};

class NativeScheduler {
 public:
  // prefix_cache=true enables refcounted shared prompt pages keyed by the
  // chained per-page hashes the Python side computes from prompt tokens
  // (semantics lockstep-pinned to engine/prefix_cache.py).
  NativeScheduler(int max_num_seqs, int max_prefill_tokens, int page_size,
                  int num_pages, int max_waiting, bool prefix_cache = false,
                  int window_tokens = 0)
      : max_num_seqs_(max_num_seqs), max_prefill_tokens_(max_prefill_tokens),
        page_size_(page_size), num_pages_(num_pages),
        max_waiting_(max_waiting), prefix_cache_(prefix_cache),
        window_tokens_(window_tokens) {
    for (int p = num_pages - 1; p >= 1; --p) free_list_.push_back(p);
  }

  bool add(long seq_id, int prompt_len,
           const std::vector<long>& page_hashes = {}) {
    if ((int)waiting_.size() >= max_waiting_) return false;
    SeqState st;
    st.prompt_len = prompt_len;
    st.num_tokens = prompt_len;
    st.hashes = page_hashes;
    seqs_[seq_id] = st;
    waiting_.push_back(seq_id);
    return true;
  }

  int num_free() const { return (int)free_list_.size(); }
  int num_queued() const { return (int)waiting_.size(); }
  int num_running() const { return (int)running_.size(); }
  long n_preempted() const { return n_preempted_; }
  bool has_work() const { return !waiting_.empty() || !running_.empty(); }

  std::vector<int> pages(long seq_id) const {
    auto it = seqs_.find(seq_id);
    return it == seqs_.end() ? std::vector<int>{} : it->second.pages;
  }

  int num_tokens(long seq_id) const {
    auto it = seqs_.find(seq_id);
    return it == seqs_.end() ? 0 : it->second.num_tokens;
  }

  void note_token(long seq_id) {  // engine appended one generated token
    auto it = seqs_.find(seq_id);
    if (it == seqs_.end()) return;
    SeqState& st = it->second;
    if (prefix_cache_ && st.num_tokens == st.prompt_len)
      register_pages(st);  // prompt fully prefilled: publish full pages
    st.num_tokens += 1;
  }

  void finish(long seq_id) {
    auto it = seqs_.find(seq_id);
    if (it == seqs_.end()) return;
    release(it->second);
    for (size_t i = 0; i < running_.size(); ++i)
      if (running_[i] == seq_id) { running_.erase(running_.begin() + i); break; }
    for (auto wi = waiting_.begin(); wi != waiting_.end(); ++wi)
      if (*wi == seq_id) { waiting_.erase(wi); break; }
    seqs_.erase(it);
  }

  int cached_prefix(long seq_id) const {
    auto it = seqs_.find(seq_id);
    return it == seqs_.end() ? 0 : it->second.cached_prefix;
  }
  long cache_hits() const { return cache_hits_; }
  long cached_tokens() const { return cached_tokens_; }
  int cache_pages() const { return (int)lru_.size(); }

  ScheduleResult schedule() {
    ScheduleResult r;
    // 1) admit prefills
    int tokens = 0;
    std::vector<long> batch;
    std::unordered_set<long> claimed;  // first-uncached hashes this batch
    while (!waiting_.empty() &&
           (int)(running_.size() + batch.size()) < max_num_seqs_) {
      long cand = waiting_.front();
      SeqState& st = seqs_[cand];
      // cached full-page prefix (>=1 prompt token always recomputed)
      std::vector<int> matched;
      int max_full = (st.prompt_len - 1) / page_size_;
      if (prefix_cache_) {
        for (int k = 0; k < (int)st.hashes.size() && k < max_full; ++k) {
          auto it = cache_.find(st.hashes[k]);
          if (it == cache_.end()) break;
          matched.push_back(it->second->second);
        }
      }
      // same-batch dedup (lockstep with the Python oracle): defer a
      // candidate whose next page another batch member will compute
      bool has_next = prefix_cache_ &&
                      (int)matched.size() < max_full &&
                      (int)matched.size() < (int)st.hashes.size();
      long nxt = has_next ? st.hashes[matched.size()] : 0;
      if (has_next && claimed.count(nxt)) break;
      int cached_tok = (int)matched.size() * page_size_;
      // num_tokens includes outputs retained across preemption — their KV
      // recomputes as prefill on re-admission
      int ntok = st.num_tokens - cached_tok;
      if (!batch.empty() && tokens + ntok > max_prefill_tokens_) break;
      // PIN matched pages before eviction can touch them (lockstep with
      // the Python oracle's fix for the same hazard)
      for (size_t k = 0; k < matched.size(); ++k) {
        refs_[matched[k]] += 1;
        lru_touch(st.hashes[k]);
      }
      int need = pages_needed(st.num_tokens) - (int)matched.size();
      if (!ensure_free(need)) {
        for (int p : matched) unref(p);  // unpin; candidate stays queued
        break;
      }
      waiting_.pop_front();
      st.pages = matched;
      alloc_into(st, need);
      st.cached_prefix = cached_tok;
      cache_hits_ += (long)matched.size();
      cached_tokens_ += cached_tok;
      if (has_next) claimed.insert(nxt);
      batch.push_back(cand);
      tokens += ntok;
    }
    if (!batch.empty()) {
      for (long s : batch) running_.push_back(s);
      r.has_work = true;
      r.is_prefill = true;
      r.seq_ids = batch;
      return r;
    }
    // 2) decode with growth + preemption
    if (running_.empty()) return r;
    size_t i = 0;
    while (i < running_.size()) {
      long sid = running_[i];
      SeqState& st = seqs_[sid];
      roll(st);
      while (!grow(st)) {
        long victim = running_.back();
        if (victim == sid) {
          running_.pop_back();
          preempt(sid, r);
          // i stays (element shifted out)
          goto next_outer;
        }
        running_.pop_back();
        preempt(victim, r);
      }
      ++i;
    next_outer:;
    }
    if (running_.empty()) return r;
    r.has_work = true;
    r.is_prefill = false;
    r.seq_ids = running_;
    return r;
  }

 private:
  int pages_needed(int ntok) const {
    return (ntok + page_size_ - 1) / page_size_;
  }

  void alloc_into(SeqState& st, int n) {
    for (int j = 0; j < n; ++j) {
      int p = free_list_.back();
      free_list_.pop_back();
      refs_[p] = 1;
      st.pages.push_back(p);
    }
  }

  // Rolling KV buffer (sliding-window models): reclaim pages wholly
  // behind the attention band.  The ID stays in st.pages so block-table
  // slots keep their position (the band mask guarantees those tokens are
  // never scored); unref handles prefix-shared pages.  The 64-token
  // slack covers attn_prefill.hip's KV-tile staging below the band.
  void roll(SeqState& st) {
    if (!window_tokens_) return;
    const int lim = st.num_tokens - window_tokens_ - 64;
    while ((st.freed_pages + 1) * page_size_ <= lim) {
      unref(st.pages[st.freed_pages]);
      st.freed_pages += 1;
    }
  }

  bool grow(SeqState& st) {
    int need = pages_needed(st.num_tokens + 1);
    if (need > (int)st.pages.size()) {
      if (!ensure_free(1)) return false;
      alloc_into(st, 1);
    }
    return true;
  }

  void release(SeqState& st) {
    for (size_t k = st.freed_pages; k < st.pages.size(); ++k)
      unref(st.pages[k]);
    st.pages.clear();
    st.freed_pages = 0;
    st.cached_prefix = 0;
  }

  void unref(int p) {
    auto it = refs_.find(p);
    if (it == refs_.end()) return;  // pre-refcount pages (never happens)
    if (--it->second == 0) {
      refs_.erase(it);
      free_list_.push_back(p);
    }
  }

  bool ensure_free(int need) {
    while ((int)free_list_.size() < need) {
      if (!evict_one()) return false;
    }
    return true;
  }

  bool evict_one() {
    // LRU-first cached page that only the cache itself references
    for (auto it = lru_.begin(); it != lru_.end(); ++it) {
      auto rit = refs_.find(it->second);
      if (rit != refs_.end() && rit->second == 1) {
        page_hash_.erase(it->second);
        cache_.erase(it->first);
        unref(it->second);
        lru_.erase(it);
        return true;
      }
    }
    return false;
  }

  void lru_touch(long h) {
    auto it = cache_.find(h);
    if (it == cache_.end()) return;
    lru_.splice(lru_.end(), lru_, it->second);  // move to MRU end
  }

  void register_pages(SeqState& st) {
    for (size_t k = 0; k < st.hashes.size() && k < st.pages.size(); ++k) {
      long h = st.hashes[k];
      if (cache_.count(h)) {
        lru_touch(h);
        continue;
      }
      lru_.emplace_back(h, st.pages[k]);
      cache_[h] = std::prev(lru_.end());
      page_hash_[st.pages[k]] = h;
      refs_[st.pages[k]] += 1;
    }
  }

  void preempt(long sid, ScheduleResult& r) {
    SeqState& st = seqs_[sid];
    release(st);
    // num_tokens keeps the generated tokens: on re-admission their KV
    // recomputes as prefill and decode resumes (vLLM-style recompute;
    // resampling would splice sampled streams)
    waiting_.push_front(sid);
    n_preempted_ += 1;
    r.preempted.push_back(sid);
  }

  int max_num_seqs_, max_prefill_tokens_, page_size_, num_pages_, max_waiting_;
  bool prefix_cache_ = false;
  int window_tokens_ = 0;
  std::vector<int> free_list_;
  std::deque<long> waiting_;
  std::vector<long> running_;
  std::unordered_map<long, SeqState> seqs_;
  long n_preempted_ = 0;
  // prefix cache state: LRU list of (hash, page); maps for O(1) lookups
  std::list<std::pair<long, int>> lru_;
  std::unordered_map<long, std::list<std::pair<long, int>>::iterator> cache_;
  std::unordered_map<int, long> page_hash_;
  std::unordered_map<int, int> refs_;
  long cache_hits_ = 0, cached_tokens_ = 0;
};

PYBIND11_MODULE(_native, m) {
  m.doc() = "agentfield_amd native runtime (ed25519 via libcrypto, scheduler)";
  m.def("ed25519_pubkey", &ed25519_pubkey);
  m.def("ed25519_sign", &ed25519_sign);
  m.def("ed25519_verify", &ed25519_verify);
  m.def("aes_gcm_encrypt", &aes_gcm_encrypt);
  m.def("aes_gcm_decrypt", &aes_gcm_decrypt);

  py::class_<ScheduleResult>(m, "ScheduleResult")
      .def_readonly("has_work", &ScheduleResult::has_work)
      .def_readonly("is_prefill", &ScheduleResult::is_prefill)
      .def_readonly("seq_ids", &ScheduleResult::seq_ids)
      .def_readonly("preempted", &ScheduleResult::preempted);

  py::class_<NativeScheduler>(m, "NativeScheduler")
      .def(py::init<int, int, int, int, int, bool, int>(),
           py::arg("max_num_seqs"),
           py::arg("max_prefill_tokens"), py::arg("page_size"),
           py::arg("num_pages"), py::arg("max_waiting") = 4096,
           py::arg("prefix_cache") = false, py::arg("window_tokens") = 0)
      .def("add", &NativeScheduler::add, py::arg("seq_id"),
           py::arg("prompt_len"),
           py::arg("page_hashes") = std::vector<long>{})
      .def("cached_prefix", &NativeScheduler::cached_prefix)
      .def("cache_hits", &NativeScheduler::cache_hits)
      .def("cached_tokens", &NativeScheduler::cached_tokens)
      .def("cache_pages", &NativeScheduler::cache_pages)
      .def("schedule", &NativeScheduler::schedule)
      .def("finish", &NativeScheduler::finish)
      .def("note_token", &NativeScheduler::note_token)
      .def("pages", &NativeScheduler::pages)
      .def("num_tokens", &NativeScheduler::num_tokens)
      .def("num_free", &NativeScheduler::num_free)
      .def("num_queued", &NativeScheduler::num_queued)
      .def("num_running", &NativeScheduler::num_running)
      .def("n_preempted", &NativeScheduler::n_preempted)
      .def("has_work", &NativeScheduler::has_work);
}
