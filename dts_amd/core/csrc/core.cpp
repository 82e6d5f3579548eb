// dts_amd native runtime core: paged-KV block manager + continuous-batching
// scheduler in C++ (SURVEY.md §2.3 rows "Paged KV-block manager with
// shared-prefix refcounting / copy-on-write" and "Continuous-batching
// scheduler" — native components; the Python twins in
// dts_amd/serving/{kv_cache,scheduler}.py define the reference semantics
// and the differential tests in tests/serving/test_core_native.py pin the
// two together).
//
// The scheduler owns sequence token/block state natively and emits each
// step's flat batch directly as torch tensors — no per-token Python work
// on the step path.

#include <torch/extension.h>

#include <cstdint>
#include <deque>
#include <list>
#include <unordered_map>
#include <vector>

namespace {

// ---------------------------------------------------------------------------
// FNV-1a-style chained hash over a token chunk
// ---------------------------------------------------------------------------
static inline uint64_t chunk_hash(uint64_t prev, const int32_t* toks, int n) {
  uint64_t h = prev * 1099511628211ULL + 14695981039346656037ULL;
  for (int i = 0; i < n; ++i) {
    h ^= (uint64_t)(uint32_t)toks[i];
    h *= 1099511628211ULL;
  }
  return h;
}

struct Block {
  int ref_count = 0;
  bool hashed = false;
  uint64_t hash = 0;
  std::vector<int32_t> content;  // token ids when full+registered
};

enum class Status : int { WAITING = 0, RUNNING = 1, FINISHED = 2, ABORTED = 3 };

static inline uint64_t bigram_key(int32_t a, int32_t b) {
  return ((uint64_t)(uint32_t)a << 32) | (uint32_t)b;
}

struct Seq {
  int64_t id;
  std::vector<int32_t> tokens;
  std::vector<int32_t> block_table;
  int num_computed = 0;
  int num_prompt = 0;
  int num_hashed = 0;
  uint64_t last_hash = 0;
  bool in_flight = false;
  int sched_chunk = 0;
  int64_t arrival = 0;
  Status status = Status::WAITING;
  uint64_t prompt_key = 0;
  uint64_t prefix_key = 0;  // hash of first min(512, len) prompt tokens
  // --- prompt-lookup speculative decoding (dts_amd/serving/spec.py is
  // the Python reference twin): bigram -> (latest, prev) continuation
  // positions over this sequence's own tokens
  std::unordered_map<uint64_t, std::pair<int, int>> ngram;
  int ngram_n = 0;
  bool allow_spec = true;
  std::vector<int32_t> draft;  // draft rows scheduled this step

  void index_tokens() {
    const int L = (int)tokens.size();
    for (int i = std::max(1, ngram_n); i < L; ++i) {
      uint64_t key = bigram_key(tokens[i - 1], tokens[i]);
      auto it = ngram.find(key);
      if (it == ngram.end())
        ngram.emplace(key, std::make_pair(i + 1, 0));
      else {
        it->second.second = it->second.first;
        it->second.first = i + 1;
      }
    }
    ngram_n = L;
  }

  std::vector<int32_t> propose(int max_k) const {
    const int L = (int)tokens.size();
    if (L < 3 || max_k <= 0) return {};
    auto it = ngram.find(bigram_key(tokens[L - 2], tokens[L - 1]));
    if (it == ngram.end()) return {};
    // the latest continuation is the tail itself (self-match) — draft
    // from the previous occurrence
    int cont = (it->second.first >= L) ? it->second.second : it->second.first;
    if (cont <= 0 || cont >= L) return {};
    int k = std::min(max_k, L - cont);
    return std::vector<int32_t>(tokens.begin() + cont, tokens.begin() + cont + k);
  }
};

class CoreScheduler {
 public:
  CoreScheduler(int num_blocks, int block_size, int64_t max_batch_tokens,
                int max_running, int spec_k = 0, int max_spec_rows = 16)
      : num_blocks_(num_blocks),
        block_size_(block_size),
        max_batch_tokens_(max_batch_tokens),
        max_running_(max_running),
        spec_k_(spec_k),
        max_spec_rows_(max_spec_rows),
        blocks_(num_blocks) {
    free_ids_.reserve(num_blocks);
    for (int i = num_blocks - 1; i >= 0; --i) free_ids_.push_back(i);
  }

  // ---- stats ------------------------------------------------------------
  int64_t cache_hit_tokens = 0;
  int64_t cache_miss_tokens = 0;
  int64_t preemptions = 0;

  int num_free() const { return (int)(free_ids_.size() + evictable_.size()); }

  // ---- sequence lifecycle ------------------------------------------------
  void add(int64_t seq_id, const std::vector<int32_t>& tokens,
           bool allow_spec = true) {
    if (tokens.empty())  // empty prompts crashed the step path (UB)
      throw py::value_error("empty prompt: at least one token required");
    Seq s;
    s.id = seq_id;
    s.tokens = tokens;
    s.num_prompt = (int)tokens.size();
    s.arrival = arrival_++;
    s.allow_spec = allow_spec;
    s.prompt_key = chunk_hash(0xABCD, tokens.data(), (int)tokens.size());
    s.prefix_key =
        chunk_hash(0xBEEF, tokens.data(), std::min<int>(512, (int)tokens.size()));
    if (spec_k_ > 0 && s.allow_spec) s.index_tokens();
    seqs_.emplace(seq_id, std::move(s));
    waiting_.push_back(seq_id);
  }

  void append_token(int64_t seq_id, int32_t tok) {
    Seq& s = seqs_.at(seq_id);
    s.tokens.push_back(tok);
    if (spec_k_ > 0 && s.allow_spec) s.index_tokens();
  }

  void extend_tokens(int64_t seq_id, const std::vector<int32_t>& toks) {
    Seq& s = seqs_.at(seq_id);
    s.tokens.insert(s.tokens.end(), toks.begin(), toks.end());
    if (spec_k_ > 0 && s.allow_spec) s.index_tokens();
  }

  void set_sched_chunk(int64_t seq_id, int chunk) {
    auto it = seqs_.find(seq_id);
    if (it != seqs_.end()) it->second.sched_chunk = chunk;
  }

  // ---- chained-decode hooks (dts_amd/serving/chain.py) ------------------
  int64_t waiting_count() const { return (int64_t)waiting_.size(); }

  bool reserve_tokens(int64_t seq_id, int upto_tokens) {
    auto it = seqs_.find(seq_id);
    if (it == seqs_.end()) return false;
    return ensure_blocks(it->second, upto_tokens);
  }

  void chain_advance(int64_t seq_id, int32_t tok) {
    auto it = seqs_.find(seq_id);
    if (it == seqs_.end()) return;
    Seq& s = it->second;
    s.tokens.push_back(tok);
    if (spec_k_ > 0 && s.allow_spec) s.index_tokens();
    s.num_computed += 1;
    register_full_blocks(s);
  }

  std::vector<int32_t> get_block_table(int64_t seq_id) const {
    auto it = seqs_.find(seq_id);
    if (it == seqs_.end()) return {};
    return it->second.block_table;
  }

  int64_t num_tokens(int64_t seq_id) const {
    return (int64_t)seqs_.at(seq_id).tokens.size();
  }

  int64_t num_computed(int64_t seq_id) const {
    return seqs_.at(seq_id).num_computed;
  }

  bool has_work() const { return !waiting_.empty() || !running_.empty(); }

  void finish(int64_t seq_id) {
    auto it = seqs_.find(seq_id);
    if (it == seqs_.end()) return;
    release_blocks(it->second);
    erase_running(seq_id);
    seqs_.erase(it);
  }

  void abort(int64_t seq_id) {
    auto it = seqs_.find(seq_id);
    if (it == seqs_.end()) return;
    release_blocks(it->second);
    erase_running(seq_id);
    for (auto w = waiting_.begin(); w != waiting_.end(); ++w)
      if (*w == seq_id) {
        waiting_.erase(w);
        break;
      }
    seqs_.erase(it);
  }

  std::vector<int64_t> take_stuck() {
    // stuck seqs are terminal: the engine fails their futures and never
    // calls finish/abort, so erase their state here or it leaks for the
    // process lifetime
    auto out = stuck_;
    stuck_.clear();
    for (int64_t sid : out) {
      auto it = seqs_.find(sid);
      if (it == seqs_.end()) continue;
      release_blocks(it->second);
      seqs_.erase(it);
    }
    return out;
  }

  // ---- the step ----------------------------------------------------------
  // returns a dict of tensors + id lists; empty dict when nothing to do
  py::dict schedule() {
    admit();
    std::vector<Seq*> prefills, decodes;
    int64_t budget = max_batch_tokens_;
    int spec_rows = 0;
    // iterate a snapshot: preempt_youngest erases victims from running_
    // mid-loop (mutating a range-for'd container is UB), and a victim
    // visited later must be skipped — growing its released block table
    // would leak blocks permanently (refcount 1, no owner)
    const std::vector<int64_t> snapshot(running_.begin(), running_.end());
    for (int64_t sid : snapshot) {
      auto sit = seqs_.find(sid);
      if (sit == seqs_.end()) continue;
      Seq& s = sit->second;
      if (s.status != Status::RUNNING || s.in_flight) continue;
      int remaining = (int)s.tokens.size() - s.num_computed;
      if (remaining <= 0) continue;
      if (!ensure_blocks(s, (int)s.tokens.size())) {
        if (!preempt_youngest(&s)) {
          if (running_.size() == 1 && prefills.empty() && decodes.empty()) {
            release_blocks(s);
            erase_running(sid);
            s.status = Status::WAITING;
            stuck_.push_back(sid);
          }
          continue;
        }
        if (s.status != Status::RUNNING) continue;  // s was the victim
        if (!ensure_blocks(s, (int)s.tokens.size())) continue;
      }
      int chunk = std::min<int64_t>(remaining, budget);
      if (chunk <= 0) continue;
      // mark immediately: a seq already placed in this step's batch must
      // not become a preemption victim for a later seq in this loop
      // (running order != arrival order after re-admission), or
      // build_batch would index its released block table
      s.in_flight = true;
      if (remaining == 1) {
        // true decode row (the tail token) — maybe add draft rows.
        // (a budget-starved chunk==1 MID-prompt is a prefill chunk;
        // classifying it as decode would sample a bogus mid-prompt token)
        s.draft.clear();
        if (spec_k_ > 0 && s.allow_spec &&
            spec_rows + 1 + spec_k_ <= max_spec_rows_ && budget > 1) {
          s.draft = s.propose((int)std::min<int64_t>(spec_k_, budget - 1));
          if (!s.draft.empty() &&
              !ensure_blocks(s, (int)s.tokens.size() + (int)s.draft.size()))
            s.draft.clear();  // never preempt for draft rows
        }
        spec_rows += 1 + (int)s.draft.size();
        budget -= 1 + (int64_t)s.draft.size();
        s.sched_chunk = 1;
        decodes.push_back(&s);
      } else {
        budget -= chunk;
        s.sched_chunk = chunk;
        prefills.push_back(&s);
      }
      if (budget <= 0) break;
    }
    if (prefills.empty() && decodes.empty()) return py::dict();
    return build_batch(prefills, decodes);
  }

  void advance() {
    // bump num_computed for everything scheduled by the last build_batch
    for (int64_t sid : last_scheduled_) {
      auto it = seqs_.find(sid);
      if (it == seqs_.end()) continue;
      Seq& s = it->second;
      s.num_computed += s.sched_chunk;
      s.in_flight = false;
      register_full_blocks(s);
    }
    last_scheduled_.clear();
  }

 private:
  // ---- block pool ---------------------------------------------------------
  int pop_free_block() {
    if (!free_ids_.empty()) {
      int b = free_ids_.back();
      free_ids_.pop_back();
      return b;
    }
    if (!evictable_.empty()) {
      int b = evictable_.front();
      evictable_.pop_front();
      evict_pos_.erase(b);
      Block& blk = blocks_[b];
      if (blk.hashed) {
        auto it = hash_table_.find(blk.hash);
        if (it != hash_table_.end() && it->second == b) hash_table_.erase(it);
        blk.hashed = false;
        blk.content.clear();
      }
      return b;
    }
    return -1;
  }

  void acquire(int b) {
    Block& blk = blocks_[b];
    if (blk.ref_count == 0) {
      auto it = evict_pos_.find(b);
      if (it != evict_pos_.end()) {
        evictable_.erase(it->second);
        evict_pos_.erase(it);
      }
    }
    blk.ref_count++;
  }

  void free_block(int b) {
    Block& blk = blocks_[b];
    TORCH_CHECK(blk.ref_count > 0, "double free of block ", b);
    if (--blk.ref_count == 0) {
      if (blk.hashed) {
        evictable_.push_back(b);
        evict_pos_[b] = std::prev(evictable_.end());
      } else {
        free_ids_.push_back(b);
      }
    }
  }

  void release_blocks(Seq& s) {
    for (int b : s.block_table) free_block(b);
    s.block_table.clear();
  }

  bool ensure_blocks(Seq& s, int upto_tokens) {
    int need = (upto_tokens + block_size_ - 1) / block_size_;
    while ((int)s.block_table.size() < need) {
      int b = pop_free_block();
      if (b < 0) return false;
      blocks_[b].ref_count = 1;
      s.block_table.push_back(b);
    }
    return true;
  }

  void register_full_blocks(Seq& s) {
    while ((s.num_hashed + 1) * block_size_ <= s.num_computed) {
      int bi = s.num_hashed;
      const int32_t* chunk = s.tokens.data() + (size_t)bi * block_size_;
      int bid = s.block_table[bi];
      Block& blk = blocks_[bid];
      if (!blk.hashed) {
        uint64_t h = chunk_hash(s.last_hash, chunk, block_size_);
        blk.hashed = true;
        blk.hash = h;
        blk.content.assign(chunk, chunk + block_size_);
        hash_table_.emplace(h, bid);  // first writer wins
        s.last_hash = h;
      } else {
        s.last_hash = blk.hash;
      }
      s.num_hashed++;
    }
  }

  // ---- prefix cache -------------------------------------------------------
  void match_prefix(Seq& s) {
    uint64_t prev = 0;
    int n = 0;
    const int total = (int)s.tokens.size();
    for (int start = 0; start + block_size_ <= total; start += block_size_) {
      uint64_t h = chunk_hash(prev, s.tokens.data() + start, block_size_);
      auto it = hash_table_.find(h);
      if (it == hash_table_.end()) break;
      Block& blk = blocks_[it->second];
      if (!blk.hashed ||
          !std::equal(blk.content.begin(), blk.content.end(),
                      s.tokens.data() + start))
        break;
      acquire(it->second);
      s.block_table.push_back(it->second);
      prev = h;
      n += block_size_;
    }
    if (n >= total) n = total - 1;  // recompute last token for its logits
    s.num_computed = n;
    s.num_hashed = (int)s.block_table.size();
    s.last_hash = prev;
  }

  // ---- admission ----------------------------------------------------------
  void admit() {
    // duplicate-prefill holdback (identical prompts of in-flight
    // prefills) + shared-prefix holdback: requests whose first 512
    // prompt tokens match an in-flight prefill wait for it, then admit
    // into a prefix-cache hit instead of re-prefilling the shared prefix
    std::unordered_map<uint64_t, int> inflight;
    std::unordered_map<uint64_t, int> inflight_prefix;
    for (int64_t sid : running_) {
      Seq& s = seqs_.at(sid);
      if (s.num_computed < s.num_prompt) {
        inflight[s.prompt_key]++;
        inflight_prefix[s.prefix_key]++;
      }
    }
    std::deque<int64_t> held;
    while (!waiting_.empty() && (int)running_.size() < max_running_) {
      int64_t sid = waiting_.front();
      waiting_.pop_front();
      Seq& s = seqs_.at(sid);
      if (inflight.count(s.prompt_key) ||
          (s.num_prompt >= 512 && inflight_prefix.count(s.prefix_key))) {
        held.push_back(sid);
        continue;
      }
      s.block_table.clear();
      match_prefix(s);
      if (!ensure_blocks(s, (int)s.tokens.size())) {
        release_blocks(s);
        s.num_computed = 0;
        s.num_hashed = 0;
        s.last_hash = 0;
        int need = ((int)s.tokens.size() + block_size_ - 1) / block_size_;
        if (need > num_blocks_) {
          stuck_.push_back(sid);
          continue;
        }
        held.push_back(sid);
        break;
      }
      cache_hit_tokens += s.num_computed;
      cache_miss_tokens += (int64_t)s.tokens.size() - s.num_computed;
      s.status = Status::RUNNING;
      running_.push_back(sid);
      if (s.num_computed < s.num_prompt) {
        inflight[s.prompt_key]++;
        inflight_prefix[s.prefix_key]++;
      }
    }
    while (!held.empty()) {
      waiting_.push_front(held.back());
      held.pop_back();
    }
  }

  bool preempt_youngest(Seq* exclude) {
    Seq* victim = nullptr;
    for (int64_t sid : running_) {
      Seq& s = seqs_.at(sid);
      if (s.in_flight) continue;
      if (!victim || s.arrival > victim->arrival) victim = &s;
    }
    if (!victim) return false;
    if (victim == exclude) {
      int others = 0;
      for (int64_t sid : running_)
        if (!seqs_.at(sid).in_flight) others++;
      if (others == 1) return false;
    }
    release_blocks(*victim);
    victim->num_computed = 0;
    victim->num_hashed = 0;
    victim->last_hash = 0;
    victim->status = Status::WAITING;
    erase_running(victim->id);
    waiting_.push_front(victim->id);
    preemptions++;
    return true;
  }

  void erase_running(int64_t sid) {
    for (auto it = running_.begin(); it != running_.end(); ++it)
      if (*it == sid) {
        running_.erase(it);
        return;
      }
  }

  // ---- batch assembly -----------------------------------------------------
  py::dict build_batch(std::vector<Seq*>& prefills, std::vector<Seq*>& decodes) {
    int64_t T = 0;
    for (Seq* s : prefills) T += s->sched_chunk;
    int64_t D_rows = 0;
    for (Seq* s : decodes) D_rows += 1 + (int64_t)s->draft.size();
    T += D_rows;

    auto opts_i64 = torch::TensorOptions().dtype(torch::kInt64);
    auto opts_i32 = torch::TensorOptions().dtype(torch::kInt32);
    torch::Tensor token_ids = torch::empty({T}, opts_i64);
    torch::Tensor positions = torch::empty({T}, opts_i64);
    torch::Tensor slots = torch::empty({T}, opts_i64);
    auto* tid = token_ids.data_ptr<int64_t>();
    auto* pos = positions.data_ptr<int64_t>();
    auto* slt = slots.data_ptr<int64_t>();

    std::vector<int64_t> sample_idx, sampled_ids, scheduled_ids, sample_pos;
    py::list row_groups;  // (seq_id, n_rows) in sampled-row order
    py::list spec;        // (seq_id, [draft tokens]) for spec'd seqs
    int64_t cursor = 0;

    const int P = (int)prefills.size();
    torch::Tensor cu_q = torch::empty({P + 1}, opts_i32);
    auto* cq = cu_q.data_ptr<int32_t>();
    cq[0] = 0;
    int max_pf_blocks = 1;
    for (Seq* s : prefills)
      max_pf_blocks = std::max(max_pf_blocks, (int)s->block_table.size());
    torch::Tensor pf_tables = torch::zeros({P, max_pf_blocks}, opts_i32);
    torch::Tensor pf_kv = torch::empty({P}, opts_i32);
    for (int i = 0; i < P; ++i) {
      Seq* s = prefills[i];
      int start = s->num_computed, end = start + s->sched_chunk;
      for (int p = start; p < end; ++p) {
        tid[cursor] = s->tokens[p];
        pos[cursor] = p;
        slt[cursor] =
            (int64_t)s->block_table[p / block_size_] * block_size_ +
            p % block_size_;
        cursor++;
      }
      cq[i + 1] = cq[i] + s->sched_chunk;
      auto row = pf_tables[i];
      auto* rp = row.data_ptr<int32_t>();
      for (size_t b = 0; b < s->block_table.size(); ++b) rp[b] = s->block_table[b];
      pf_kv.data_ptr<int32_t>()[i] = end;
      if (end == (int)s->tokens.size()) {
        sample_idx.push_back(cursor - 1);
        sampled_ids.push_back(s->id);
        sample_pos.push_back((int64_t)s->tokens.size());
        row_groups.append(py::make_tuple(s->id, 1));
      }
      s->in_flight = true;
      scheduled_ids.push_back(s->id);
    }
    int64_t num_prefill_tokens = cursor;

    int max_dc_blocks = 1;
    for (Seq* s : decodes)
      max_dc_blocks = std::max(max_dc_blocks, (int)s->block_table.size());
    torch::Tensor dc_tables = torch::zeros({D_rows, max_dc_blocks}, opts_i32);
    torch::Tensor dc_kv = torch::empty({D_rows}, opts_i32);
    int64_t row = 0;
    for (Seq* s : decodes) {
      const int p = s->num_computed;
      const int k = (int)s->draft.size();
      // row 0 = the real tail token; rows 1..k = draft candidates, each
      // an independent 1-token decode row with its own position/kv_len
      // (rope_kv_append writes all rows' KV before attention runs)
      for (int j = 0; j <= k; ++j) {
        const int q = p + j;
        tid[cursor] = (j == 0) ? s->tokens[p] : s->draft[j - 1];
        pos[cursor] = q;
        slt[cursor] =
            (int64_t)s->block_table[q / block_size_] * block_size_ +
            q % block_size_;
        auto* rp = dc_tables[row].data_ptr<int32_t>();
        for (size_t b = 0; b < s->block_table.size(); ++b)
          rp[b] = s->block_table[b];
        dc_kv.data_ptr<int32_t>()[row] = q + 1;
        sample_idx.push_back(cursor);
        sampled_ids.push_back(s->id);
        sample_pos.push_back((int64_t)s->tokens.size() + j);
        cursor++;
        row++;
      }
      row_groups.append(py::make_tuple(s->id, k + 1));
      if (k) {
        py::list dl;
        for (int32_t d : s->draft) dl.append((int64_t)d);
        spec.append(py::make_tuple(s->id, dl));
      }
      s->in_flight = true;
      scheduled_ids.push_back(s->id);
    }
    last_scheduled_ = scheduled_ids;

    py::dict out;
    out["token_ids"] = token_ids;
    out["positions"] = positions;
    out["slot_mapping"] = slots;
    out["num_prefill_seqs"] = P;
    out["num_prefill_tokens"] = num_prefill_tokens;
    if (P) {
      out["cu_q"] = cu_q;
      out["prefill_block_tables"] = pf_tables;
      out["prefill_kv_lens"] = pf_kv;
    }
    out["num_decode_seqs"] = D_rows;  // decode ROWS (>= seqs with spec on)
    if (D_rows) {
      out["decode_block_tables"] = dc_tables;
      out["decode_kv_lens"] = dc_kv;
    }
    out["sample_indices"] =
        torch::tensor(sample_idx, opts_i64);
    out["sampled_ids"] = sampled_ids;
    out["scheduled_ids"] = scheduled_ids;
    out["sample_pos"] = sample_pos;
    out["row_groups"] = row_groups;
    out["spec"] = spec;
    return out;
  }

  int num_blocks_;
  int block_size_;
  int64_t max_batch_tokens_;
  int max_running_;
  int spec_k_;
  int max_spec_rows_;
  std::vector<Block> blocks_;
  std::vector<int> free_ids_;
  std::list<int> evictable_;
  std::unordered_map<int, std::list<int>::iterator> evict_pos_;
  std::unordered_map<uint64_t, int> hash_table_;
  std::unordered_map<int64_t, Seq> seqs_;
  std::deque<int64_t> waiting_;
  std::vector<int64_t> running_;
  std::vector<int64_t> stuck_;
  std::vector<int64_t> last_scheduled_;
  int64_t arrival_ = 0;
};

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  py::class_<CoreScheduler>(m, "CoreScheduler")
      .def(py::init<int, int, int64_t, int, int, int>(), py::arg("num_blocks"),
           py::arg("block_size"), py::arg("max_batch_tokens"),
           py::arg("max_running"), py::arg("spec_k") = 0,
           py::arg("max_spec_rows") = 16)
      .def("add", &CoreScheduler::add, py::arg("seq_id"), py::arg("tokens"),
           py::arg("allow_spec") = true)
      .def("set_sched_chunk", &CoreScheduler::set_sched_chunk)
      .def("waiting_count", &CoreScheduler::waiting_count)
      .def("reserve_tokens", &CoreScheduler::reserve_tokens)
      .def("chain_advance", &CoreScheduler::chain_advance)
      .def("get_block_table", &CoreScheduler::get_block_table)
      .def("append_token", &CoreScheduler::append_token)
      .def("extend_tokens", &CoreScheduler::extend_tokens)
      .def("num_tokens", &CoreScheduler::num_tokens)
      .def("num_computed", &CoreScheduler::num_computed)
      .def("schedule", &CoreScheduler::schedule)
      .def("advance", &CoreScheduler::advance)
      .def("finish", &CoreScheduler::finish)
      .def("abort", &CoreScheduler::abort)
      .def("has_work", &CoreScheduler::has_work)
      .def("take_stuck", &CoreScheduler::take_stuck)
      .def("num_free", &CoreScheduler::num_free)
      .def_readonly("cache_hit_tokens", &CoreScheduler::cache_hit_tokens)
      .def_readonly("cache_miss_tokens", &CoreScheduler::cache_miss_tokens)
      .def_readonly("preemptions", &CoreScheduler::preemptions);
}
