// Plain-C function table handed from kungfu_amd._core to kungfu_amd._rccl
// through a PyCapsule, so the RCCL layer can bootstrap communicators over
// the framework's OWN control plane (reference semantics:
// srcs/cpp/src/nccl/gpu_collective.cpp:169-191 — the 128-byte ncclUniqueId
// is broadcast over the CPU collective, never over a second rendezvous
// system like a TCPStore) and broadcast scheduler order agreements
// (scheduler.cpp:93-119).
//
// The table is plain C so the two extensions can be built by different
// compilers (g++ for _core, hipcc for _rccl) without ABI coupling.
#pragma once

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

#define KF_CONTROL_API_VERSION 1
#define KF_CONTROL_API_CAPSULE "kungfu_amd._core.control_api"

typedef struct kf_control_api {
    uint32_t api_version;  // KF_CONTROL_API_VERSION
    void *ctx;             // opaque (kf::Peer*)

    int (*rank)(void *ctx);
    int (*size)(void *ctx);
    int (*local_rank)(void *ctx);
    int (*local_size)(void *ctx);
    int (*host_count)(void *ctx);
    // index of this peer's host among distinct hosts (first-seen order);
    // the rank of the local master inside the cross-host scope
    int (*host_rank)(void *ctx);
    uint32_t (*cluster_version)(void *ctx);

    // byte-buffer collectives over the CPU/TCP collective engine
    // (blocking; in-place on buf). root is a GLOBAL rank.
    void (*broadcast)(void *ctx, void *buf, size_t len, int root,
                      const char *name);
    // intra-host broadcast rooted at the local master
    void (*local_broadcast)(void *ctx, void *buf, size_t len,
                            const char *name);
    void (*barrier)(void *ctx);
} kf_control_api;

#ifdef __cplusplus
}
#endif
