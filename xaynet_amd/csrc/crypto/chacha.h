// ChaCha20 keystream + a byte-stream reader that reproduces Rust
// rand_chacha::ChaCha20Rng::fill_bytes exactly.
//
// The reference mask PRNG (rust/xaynet-core/src/crypto/prng.rs:16-27) calls
// ChaCha20Rng::from_seed(seed) and then fill_bytes(len) repeatedly. rand_core's
// BlockRng consumes the keystream in 4-byte *word* granularity: a fill of n
// bytes consumes ceil(n/4) keystream words and discards the trailing
// 4*ceil(n/4)-n bytes of the last word. Bit-exact mask parity depends on this.
//
// The keystream itself is the DJB/IETF ChaCha20 stream with key = seed,
// nonce = 0, block counter starting at 0 (both variants agree while the
// counter fits 32 bits, which it always does here per call).
#pragma once

#include <cstring>

#include <cstdint>
#include <cstddef>

namespace xaynet::crypto {

// One 64-byte ChaCha20 block. counter is the block index.
void chacha20_block(const uint8_t key[32], uint64_t counter, const uint8_t nonce[12],
                    uint8_t out[64]);

class ChaChaRng {
  public:
    explicit ChaChaRng(const uint8_t seed[32]);

    // rand_core BlockRng::fill_bytes semantics (word-granular consumption).
    void fill_bytes(uint8_t* out, size_t n);

    // hot path for the rejection sampler: one draw value of `nbytes` (<= 8)
    // little-endian bytes, consuming whole words — identical stream to
    // fill_bytes, without the per-call copy/loop overhead
    inline uint64_t draw_u64(int nbytes) {
        if (block_off_ + size_t(nbytes) > BUF) return draw_u64_slow(nbytes);
        uint64_t v = 0;
        std::memcpy(&v, buf_ + block_off_, 8 <= int(BUF - block_off_) ? 8 : nbytes);
        if (nbytes < 8) v &= (~uint64_t(0)) >> (8 * (8 - nbytes));
        size_t words = size_t(nbytes + 3) / 4;
        block_off_ += words * 4;
        word_pos_ += words;
        return v;
    }

    // Total keystream words consumed so far (diagnostics / GPU parity tests).
    uint64_t words_consumed() const { return word_pos_; }

  private:
    static constexpr size_t BUF = 256;  // 4 blocks per refill (vectorizable)
    void refill();
    uint64_t draw_u64_slow(int nbytes);

    uint8_t key_[32];
    uint8_t buf_[BUF];
    uint64_t block_idx_ = 0;   // next block index to generate
    uint64_t word_pos_ = 0;    // global word position consumed
    size_t block_off_ = BUF;   // byte offset into buf_ (word aligned), BUF = empty
};

}  // namespace xaynet::crypto
