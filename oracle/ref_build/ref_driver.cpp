// Driver for the compiled-reference oracle (oracle/_ref/conflux_ref).
//
// Compiles the reference's own LU loop (conflux_opt.hpp / conflux_opt.cpp,
// from /root/reference, sources untouched) with CONFLUX_WITH_VALIDATION so
// LU_rep fills the result buffer C (factored matrix, pivoted row order,
// tile-cyclic layout) and the permutation vector (conflux_opt.hpp:1821-1823),
// then dumps them to raw files for the Python parity tests.
//
// Usage:
//   mpiexec -n P ./conflux_ref N v Px Py Pz <input.bin|-> <outprefix> [reps]
//
// input.bin: full N*N global row-major fp64 matrix, read by every rank,
//            scattered by the owner map of layout.cpp:95-123.  "-" keeps the
//            reference's own InitMatrix fill (hard-coded KATs for
//            N in {8,9,16,20,27,32}).
// outputs:   <outprefix>.perm   : M int32  (rank 0)
//            <outprefix>.A.rK   : Ml*Nl fp64 input local buffer  (pk==0 ranks)
//            <outprefix>.C.rK   : Ml*Nl fp64 factored local buffer (pk==0)
//            <outprefix>.time   : best wall-ms over reps (rank 0, text)
#include <conflux/lu/conflux_opt.hpp>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

static void dump(const std::string &path, const void *p, std::size_t bytes) {
    FILE *f = fopen(path.c_str(), "wb");
    if (!f) { perror(path.c_str()); MPI_Abort(MPI_COMM_WORLD, 2); }
    fwrite(p, 1, bytes, f);
    fclose(f);
}

int main(int argc, char **argv) {
    MPI_Init(&argc, &argv);
    if (argc < 8) {
        std::fprintf(stderr, "usage: %s N v Px Py Pz <input.bin|-> outprefix [reps]\n", argv[0]);
        MPI_Abort(MPI_COMM_WORLD, 1);
    }
    int N = std::atoi(argv[1]), v = std::atoi(argv[2]);
    int Px = std::atoi(argv[3]), Py = std::atoi(argv[4]), Pz = std::atoi(argv[5]);
    std::string input = argv[6], outprefix = argv[7];
    int reps = argc > 8 ? std::atoi(argv[8]) : 1;

    {  // scope: ~lu_params frees its comms, which must precede MPI_Finalize
    conflux::lu_params<double> params(N, N, v, Px, Py, Pz, MPI_COMM_WORLD);

    // Optionally override the reference's per-rank RNG fill with a shared
    // global input so the parity comparison is on one well-defined matrix
    // (the reference's per-rank stream is grid-dependent).
    std::vector<double> Aglob;
    if (input != "-") {
        Aglob.resize((std::size_t)params.N * params.N);
        FILE *f = fopen(input.c_str(), "rb");
        if (!f) { perror(input.c_str()); MPI_Abort(MPI_COMM_WORLD, 2); }
        if (fread(Aglob.data(), sizeof(double), Aglob.size(), f) != Aglob.size()) {
            std::fprintf(stderr, "short read on %s\n", input.c_str());
            MPI_Abort(MPI_COMM_WORLD, 2);
        }
        fclose(f);
    }

    auto fill_local = [&]() {
        params.InitMatrix();
        if (Aglob.empty()) return;
        if (params.pk != 0) {
            std::fill(params.data.begin(), params.data.end(), 0.0);
            return;
        }
        // owner map of layout.cpp:95-123
        for (int lti = 0; lti < params.tA11x; ++lti) {
            int gti = lti * params.Px + params.pi;
            for (int ltj = 0; ltj < params.tA11y; ++ltj) {
                int gtj = ltj * params.Py + params.pj;
                for (int r = 0; r < v; ++r) {
                    const double *src = &Aglob[(std::size_t)(gti * v + r) * params.N + gtj * v];
                    double *dst = &params.data[(std::size_t)(lti * v + r) * params.Nl + ltj * v];
                    std::memcpy(dst, src, sizeof(double) * v);
                }
            }
        }
    };

    std::vector<double> C(params.data.size());
    std::vector<int> perm(params.M);
    std::size_t best_ms = ~0ull;
    for (int rep = 0; rep < reps; ++rep) {
        fill_local();
        std::size_t ms = conflux::LU_rep<double>(params, C.data(), perm.data());
        if (ms < best_ms) best_ms = ms;
        if (rep == 0) {  // dump the first repetition's results
            char suff[32];
            std::snprintf(suff, sizeof suff, ".r%d", params.rank);
            if (params.pk == 0) {
                fill_local();  // re-create the input for the .A dump
                dump(outprefix + ".A" + suff, params.data.data(),
                     params.data.size() * sizeof(double));
                fill_local();
                // (LU_rep factors a copy; params.data is not clobbered --
                //  conflux_opt.hpp:398 -- but re-fill for determinism anyway)
                dump(outprefix + ".C" + suff, C.data(), C.size() * sizeof(double));
            }
            if (params.rank == 0)
                dump(outprefix + ".perm", perm.data(), perm.size() * sizeof(int));
        }
    }
    if (params.rank == 0) {
        FILE *f = fopen((outprefix + ".time").c_str(), "w");
        std::fprintf(f, "%zu\n", best_ms);
        fclose(f);
        std::printf("_result_ lu,conflux-ref,%d,%d,%d,%dx%dx%d,time,other,%zu,%d\n",
                    params.N, params.N, params.P, Px, Py, Pz, best_ms, v);
    }
    }  // ~lu_params
    MPI_Finalize();
    return 0;
}
