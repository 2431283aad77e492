// HIP hashing vectorizer — CDNA4 (gfx950) kernels.
//
// Device replacement for the murmurhash3/CSR core of sklearn's
// HashingVectorizer that the reference's Encoderizer text pipelines call
// (reference skdist/preprocessing.py:264-310, _defaults.py:91-198;
// SURVEY.md §2.4 "HashingVectorizer" row).  The documents travel to HBM
// ONCE as a packed byte buffer; tokenization + murmur3 + feature hashing
// run on-device and emit (doc*F + feature, ±1) COO pairs through a
// global bump allocator; the torch side sorts + segment-sums them into a
// CSR (skdist_amd/ops/__init__.py:hash_vectorize).
//
// Analyzer parity with sklearn (exact for ASCII input, which the host
// wrapper verifies before taking this path):
//   * word:    token_pattern \w\w+ (runs of [A-Za-z0-9_], length >= 2),
//              n-grams joined by single spaces;
//   * char_wb: whitespace-split words padded with one space each side,
//              sliding char n-grams; a word shorter than n yields the
//              whole padded word once (sklearn's offset==0 break).
// Hash: MurmurHash3_x86_32(token_bytes, seed=0), signed;
//   index = abs(h) % n_features, value = +1 / -1 (alternate_sign).
#include "common.h"

#define GRAM_MAX 768

static __device__ __forceinline__ unsigned rotl32(unsigned x, int r) {
    return (x << r) | (x >> (32 - r));
}

static __device__ int murmur3_32(const unsigned char* data, int len,
                                 unsigned seed) {
    const unsigned c1 = 0xcc9e2d51u, c2 = 0x1b873593u;
    unsigned h1 = seed;
    const int nblocks = len / 4;
    for (int i = 0; i < nblocks; ++i) {
        unsigned k1 = (unsigned)data[i * 4] |
                      ((unsigned)data[i * 4 + 1] << 8) |
                      ((unsigned)data[i * 4 + 2] << 16) |
                      ((unsigned)data[i * 4 + 3] << 24);
        k1 *= c1;
        k1 = rotl32(k1, 15);
        k1 *= c2;
        h1 ^= k1;
        h1 = rotl32(h1, 13);
        h1 = h1 * 5u + 0xe6546b64u;
    }
    unsigned k1 = 0;
    switch (len & 3) {
        case 3: k1 ^= (unsigned)data[nblocks * 4 + 2] << 16;
        case 2: k1 ^= (unsigned)data[nblocks * 4 + 1] << 8;
        case 1:
            k1 ^= (unsigned)data[nblocks * 4];
            k1 *= c1;
            k1 = rotl32(k1, 15);
            k1 *= c2;
            h1 ^= k1;
    }
    h1 ^= (unsigned)len;
    h1 ^= h1 >> 16;
    h1 *= 0x85ebca6bu;
    h1 ^= h1 >> 13;
    h1 *= 0xc2b2ae35u;
    h1 ^= h1 >> 16;
    return (int)h1;
}

static __device__ __forceinline__ void
emit(unsigned long long doc, const unsigned char* buf, int len, unsigned F,
     int alt_sign, unsigned long long* keys, float* vals,
     unsigned long long* ctr, long long cap) {
    const int h = murmur3_32(buf, len, 0u);
    const unsigned idx =
        (h < 0 ? (unsigned)(-(long long)h) : (unsigned)h) % F;
    const float v = (!alt_sign || h >= 0) ? 1.f : -1.f;
    const unsigned long long pos = atomicAdd(ctr, 1ull);
    if (pos < (unsigned long long)cap) {
        keys[pos] = doc * (unsigned long long)F + idx;
        vals[pos] = v;
    }
}

static __device__ __forceinline__ bool is_word_char(unsigned char c) {
    // ASCII \w; bytes >= 0x80 (utf-8 continuation/lead) are treated as
    // word chars — exact for pure-ASCII docs (host gates on that)
    return (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z') ||
           (c >= '0' && c <= '9') || c == '_' || c >= 0x80;
}

static __device__ __forceinline__ bool is_space(unsigned char c) {
    return c == ' ' || c == '\t' || c == '\n' || c == '\r' || c == '\f' ||
           c == '\v';
}

#define MAX_NGRAM_TOKENS 8

extern "C" __global__ __launch_bounds__(256) void k_hash_vectorize(
    const unsigned char* __restrict__ bytes,
    const long long* __restrict__ doc_off,  // [n_docs + 1]
    long long n_docs, int mode,             // 0 = word, 1 = char_wb
    int min_n, int max_n, int n_features, int alt_sign,
    unsigned long long* __restrict__ out_keys,
    float* __restrict__ out_vals, unsigned long long* __restrict__ ctr,
    long long cap) {
    const unsigned F = (unsigned)n_features;
    for (long long d = (long long)blockIdx.x * blockDim.x + threadIdx.x;
         d < n_docs; d += (long long)gridDim.x * blockDim.x) {
        const long long lo = doc_off[d], hi = doc_off[d + 1];
        if (mode == 0) {
            // ---- word analyzer ----
            long long tok_start[MAX_NGRAM_TOKENS];
            int tok_len[MAX_NGRAM_TOKENS];
            int n_tok = 0;
            long long p = lo;
            while (p < hi) {
                while (p < hi && !is_word_char(bytes[p])) ++p;
                const long long s = p;
                while (p < hi && is_word_char(bytes[p])) ++p;
                const int len = (int)(p - s);
                if (len < 2) continue;  // token_pattern \w\w+
                // shift ring
                if (n_tok == MAX_NGRAM_TOKENS) {
                    for (int i = 1; i < MAX_NGRAM_TOKENS; ++i) {
                        tok_start[i - 1] = tok_start[i];
                        tok_len[i - 1] = tok_len[i];
                    }
                    --n_tok;
                }
                tok_start[n_tok] = s;
                tok_len[n_tok] = len;
                ++n_tok;
                // n-grams ending at this token
                unsigned char gram[GRAM_MAX];
                for (int n = min_n; n <= max_n && n <= n_tok; ++n) {
                    if (n == 1) {
                        // unigram: hash straight from the global buffer
                        emit((unsigned long long)d, bytes + s, len, F,
                             alt_sign, out_keys, out_vals, ctr, cap);
                        continue;
                    }
                    int glen = 0;
                    bool ok = true;
                    for (int t = n_tok - n; t < n_tok; ++t) {
                        if (glen + tok_len[t] + 1 > GRAM_MAX) {
                            ok = false;
                            break;
                        }
                        if (glen) gram[glen++] = ' ';
                        for (int b = 0; b < tok_len[t]; ++b)
                            gram[glen++] = bytes[tok_start[t] + b];
                    }
                    if (ok)
                        emit((unsigned long long)d, gram, glen, F,
                             alt_sign, out_keys, out_vals, ctr, cap);
                }
            }
        } else {
            // ---- char_wb analyzer ----
            long long p = lo;
            unsigned char win[64];
            while (p < hi) {
                while (p < hi && is_space(bytes[p])) ++p;
                const long long s = p;
                while (p < hi && !is_space(bytes[p])) ++p;
                const int wlen = (int)(p - s);
                if (wlen == 0) continue;
                const int plen = wlen + 2;  // ' ' + word + ' '
                for (int n = min_n; n <= max_n; ++n) {
                    if (n >= plen) {
                        // short word: whole padded word, once
                        win[0] = ' ';
                        for (int b = 0; b < wlen && b < 62; ++b)
                            win[b + 1] = bytes[s + b];
                        win[wlen + 1] = ' ';
                        emit((unsigned long long)d, win, plen, F,
                             alt_sign, out_keys, out_vals, ctr, cap);
                        break;
                    }
                    for (int off = 0; off + n <= plen; ++off) {
                        for (int b = 0; b < n; ++b) {
                            const int q = off + b;
                            win[b] = (q == 0 || q == plen - 1)
                                         ? (unsigned char)' '
                                         : bytes[s + q - 1];
                        }
                        emit((unsigned long long)d, win, n, F, alt_sign,
                             out_keys, out_vals, ctr, cap);
                    }
                }
            }
        }
    }
}

extern "C" hipError_t skdist_hash_vectorize(
    const void* bytes, const void* doc_off, long long n_docs, int mode,
    int min_n, int max_n, int n_features, int alt_sign, void* out_keys,
    void* out_vals, void* ctr, long long cap, hipStream_t stream) {
    const int blocks =
        (int)std::min<long long>((n_docs + 255) / 256, 16384);
    hipLaunchKernelGGL(k_hash_vectorize, dim3(blocks), dim3(256), 0,
                       stream, (const unsigned char*)bytes,
                       (const long long*)doc_off, n_docs, mode, min_n,
                       max_n, n_features, alt_sign,
                       (unsigned long long*)out_keys, (float*)out_vals,
                       (unsigned long long*)ctr, cap);
    return hipGetLastError();
}
