// Native paged-KV block allocator (the C++ runtime twin of
// engine/kvcache._PyBlockAllocator — identical LIFO semantics, bound into
// Python via the torch extension). Replaces the reference's per-request
// connection bookkeeping (request_handler.py:15) as the managed resource.
#include <stdexcept>
#include <vector>

class BlockAllocator {
  public:
    explicit BlockAllocator(int64_t num_blocks) : num_blocks_(num_blocks) {
        free_.reserve(num_blocks);
        for (int64_t i = num_blocks - 1; i >= 0; --i) free_.push_back(i);
    }

    int64_t num_free() const { return static_cast<int64_t>(free_.size()); }

    std::vector<int64_t> allocate(int64_t n) {
        if (n > num_free()) {
            throw std::runtime_error(
                "KV cache out of blocks: need " + std::to_string(n) + ", have " +
                std::to_string(num_free()));
        }
        std::vector<int64_t> out;
        out.reserve(n);
        for (int64_t i = 0; i < n; ++i) {
            out.push_back(free_.back());
            free_.pop_back();
        }
        return out;
    }

    void free_blocks(const std::vector<int64_t>& blocks) {
        for (auto it = blocks.rbegin(); it != blocks.rend(); ++it) {
            free_.push_back(*it);
        }
    }

    int64_t capacity() const { return num_blocks_; }

  private:
    int64_t num_blocks_;
    std::vector<int64_t> free_;
};
