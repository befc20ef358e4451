// Minimal stand-in for the COSTA library (the reference's libs/costa git
// submodule is NOT vendored in /root/reference — empty dir, version unpinned;
// see SURVEY.md §2 + §8c).  This header provides exactly the surface the
// reference's lu_params.hpp / layout.cpp use so the reference LU loop can be
// compiled in-container as the parity oracle (oracle/_ref).
//
// Semantics implemented from the owner map the reference itself specifies in
// src/conflux/lu/layout.cpp:63-135: each local block carries a pointer into
// the rank-local buffer, its leading dimension and its global block coords;
// initialize(f) fills every local block element (r, c) with
// f(row_split[block.row] + r, col_split[block.col] + c), blocks visited in
// the order layout.cpp registered them, elements row-major.  (The traversal
// order only matters for the reference's per-rank sequential mt19937 fill,
// which the parity runs bypass by overwriting lu_params::data afterwards.)
#pragma once
#include <cassert>
#include <functional>
#include <vector>

namespace costa {

struct block_t {
    void *data = nullptr;
    int ld = 0;
    int row = 0;  // global block-row coordinate
    int col = 0;  // global block-col coordinate
};

template <typename T>
class grid_layout {
   public:
    struct local_block {
        T *data;
        int ld;
        int row_off, col_off;  // global element offsets of this block
        int n_rows, n_cols;
    };
    std::vector<local_block> blocks;

    grid_layout() = default;

    void initialize(std::function<T(int, int)> f) {
        for (auto &b : blocks) {
            for (int r = 0; r < b.n_rows; ++r)
                for (int c = 0; c < b.n_cols; ++c)
                    b.data[r * (std::size_t)b.ld + c] = f(b.row_off + r, b.col_off + c);
        }
    }
};

// Signature shape of costa::custom_layout as called at
// src/conflux/lu/layout.cpp:125-133.
template <typename T>
grid_layout<T> custom_layout(int n_block_rows, int n_block_cols,
                             const int *row_splits, const int *col_splits,
                             const int * /*owners*/, int n_local_blocks,
                             const block_t *local_blocks, char ordering) {
    assert(ordering == 'R' || ordering == 'r');
    grid_layout<T> g;
    g.blocks.reserve(n_local_blocks);
    for (int i = 0; i < n_local_blocks; ++i) {
        const block_t &b = local_blocks[i];
        assert(b.row >= 0 && b.row < n_block_rows);
        assert(b.col >= 0 && b.col < n_block_cols);
        typename grid_layout<T>::local_block lb;
        lb.data = static_cast<T *>(b.data);
        lb.ld = b.ld;
        lb.row_off = row_splits[b.row];
        lb.col_off = col_splits[b.col];
        lb.n_rows = row_splits[b.row + 1] - row_splits[b.row];
        lb.n_cols = col_splits[b.col + 1] - col_splits[b.col];
        g.blocks.push_back(lb);
    }
    return g;
}

// Signature shape of costa::block_cyclic_layout as called at
// src/conflux/lu/layout.cpp:48-58.  Only instantiated (never called) by the
// oracle driver path, which goes through the MPI_Comm overload of
// conflux_layout -> custom_layout.
template <typename T>
grid_layout<T> block_cyclic_layout(int, int, int, int, int, int, int, int,
                                   int, int, char, int, int, T *, int, char,
                                   int) {
    return grid_layout<T>{};
}

}  // namespace costa
