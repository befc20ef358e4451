// selfspawn.hpp — shared launch topology for the miniapp CLIs.
//
// Multi-rank runs self-spawn one fresh process per GPU via fork+EXEC (ROCm
// HIP is not fork-safe after runtime init, so the parent makes NO HIP calls
// at all; each child re-enters main with CONFLUX_RANK set and exactly one
// GPU visible).  The RCCL unique id travels through a temp file that rank 0
// publishes atomically (sidecar write + rename).  External launchers set
// CONFLUX_RANK / CONFLUX_WORLD / CONFLUX_UID_FILE themselves.
#pragma once
#include <hip/hip_runtime.h>
#include <sys/wait.h>
#include <unistd.h>

#include <algorithm>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <functional>
#include <string>
#include <vector>

#include "../../include/conflux_lu.h"

// (Px,Py,Pz) from P, the reference's square-matrix heuristic
// (lu_params.hpp:21-47 get_p_grid with M == N, i.e. ratio == 1)
static inline void conflux_grid_from_P(int P, int *Px, int *Py, int *Pz) {
    const int p1 = (int)std::cbrt((double)P);
    const int psq = (int)std::sqrt((double)P);
    const int phsq = (int)std::sqrt((double)P / 2.0);
    if (psq * psq == P) { *Px = *Py = psq; *Pz = 1; return; }
    if (phsq * phsq == P / 2) { *Px = *Py = phsq; *Pz = 2; return; }
    int d[3] = {p1, p1, P / std::max(1, p1 * p1)};
    std::sort(d, d + 3, std::greater<int>());
    *Px = d[0]; *Py = d[1]; *Pz = d[2];
}

// GPU count probed in a throwaway CHILD so the parent never initializes the
// HIP runtime.
static inline int conflux_probe_gpu_count() {
    pid_t pid = fork();
    if (pid == 0) {
        int n = 0;
        if (hipGetDeviceCount(&n) != hipSuccess) n = 0;
        _exit(n < 0 ? 0 : (n & 0x7f));
    }
    int status = 0;
    if (pid < 0 || waitpid(pid, &status, 0) < 0 || !WIFEXITED(status))
        return 0;
    return WEXITSTATUS(status);
}

// Child path of an env-launched rank: fill `uid` from CONFLUX_UID_FILE.
// Rank 0 of a self-spawned job (CONFLUX_UID_WRITE set) generates and
// publishes it; others poll for the rename.  Returns 0 on success.
static inline int conflux_resolve_uid(char *uid) {
    const char *uf = std::getenv("CONFLUX_UID_FILE");
    if (!uf) return 1;
    if (std::getenv("CONFLUX_UID_WRITE")) {
        if (conflux_lu_make_uid(uid)) return 1;
        const std::string tmp = std::string(uf) + ".w";
        FILE *f = std::fopen(tmp.c_str(), "wb");
        if (!f ||
            std::fwrite(uid, 1, CONFLUX_LU_UID_BYTES, f) !=
                CONFLUX_LU_UID_BYTES) {
            std::fprintf(stderr, "cannot write uid file\n");
            return 1;
        }
        std::fclose(f);
        if (std::rename(tmp.c_str(), uf)) { perror("rename"); return 1; }
        return 0;
    }
    for (int t = 0; t < 6000; ++t) {  // up to ~60 s for rank 0's rename
        FILE *f = std::fopen(uf, "rb");
        if (f) {
            const bool got =
                std::fread(uid, 1, CONFLUX_LU_UID_BYTES, f) ==
                CONFLUX_LU_UID_BYTES;
            std::fclose(f);
            if (got) return 0;
        }
        usleep(10000);
    }
    std::fprintf(stderr, "timed out waiting for uid file\n");
    return 1;
}

// Parent path: spawn P fresh children of this binary (argv + extra_arg, so
// an auto-derived grid reaches every rank) and exit with their combined
// status.  Never returns.
static inline void conflux_selfspawn(int P, int argc, char **argv,
                                     const std::string &extra_arg) {
    // CONFLUX_SPAWN_OVERSUBSCRIBE=1: allow more ranks than GPUs, all
    // sharing device 0 — only meaningful under the shimccl test transport
    // (real RCCL refuses duplicate GPUs)
    const bool oversub = std::getenv("CONFLUX_SPAWN_OVERSUBSCRIBE") != nullptr;
    const int ndev = oversub ? P : conflux_probe_gpu_count();
    if (ndev < P) {
        std::fprintf(stderr,
                     "[conflux] %d GPUs visible but grid needs %d "
                     "(use --sim for single-GPU validation)\n",
                     ndev, P);
        std::exit(1);
    }
    char uidpath[] = "/tmp/conflux_uid_XXXXXX";
    {
        int fd = mkstemp(uidpath);
        if (fd < 0) { perror("mkstemp"); std::exit(1); }
        close(fd);
        unlink(uidpath);  // rank 0 re-creates it by rename
    }
    std::vector<pid_t> pids(P);
    for (int r = 0; r < P; ++r) {
        pid_t pid = fork();
        if (pid == 0) {
            char rbuf[16], wbuf[16];
            std::snprintf(rbuf, sizeof rbuf, "%d", r);
            std::snprintf(wbuf, sizeof wbuf, "%d", P);
            setenv("CONFLUX_RANK", rbuf, 1);
            setenv("CONFLUX_WORLD", wbuf, 1);
            if (!oversub) setenv("HIP_VISIBLE_DEVICES", rbuf, 1);
            setenv("CONFLUX_UID_FILE", uidpath, 1);
            if (r == 0) setenv("CONFLUX_UID_WRITE", "1", 1);
            std::vector<char *> nargv(argv, argv + argc);
            nargv.push_back(const_cast<char *>(extra_arg.c_str()));
            nargv.push_back(nullptr);
            execv("/proc/self/exe", nargv.data());
            perror("execv");
            _exit(127);
        }
        pids[r] = pid;
    }
    int status = 0, bad = 0;
    for (int r = 0; r < P; ++r) {
        waitpid(pids[r], &status, 0);
        if (!WIFEXITED(status) || WEXITSTATUS(status)) bad = 1;
    }
    unlink(uidpath);
    std::exit(bad);
}
