/* caffe_amd.h — C-ABI of the MI355X-native Caffe-MPI training engine.
 *
 * Each entry point names the reference interface it replaces (file:line in
 * the Caffe-MPI tree).  The reference exposes no C ABI of its own (pycaffe
 * is boost::python); this ABI is the FFI a maintainer would bind to reach
 * the same Layer/Net/Solver surface — see INTEGRATION.md for the ctypes
 * binding the in-repo python mirror uses.
 *
 * Conventions: plain pointers + sizes, fp32 host buffers, 0 = success /
 * nonzero = error (last message via caffe_last_error).  No torch types.
 */
#ifndef CAFFE_AMD_H_
#define CAFFE_AMD_H_

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef void* caffe_solver_t; /* caffe::Solver,  solver.hpp:73 */
typedef void* caffe_net_t;    /* caffe::Net,     net.hpp */

const char* caffe_last_error(void);

/* Caffe::set_mode / SetDevice (common.hpp): mode 0 = CPU, 1 = GPU */
int caffe_set_mode(int mode, int device);
/* GEMM compute dtype: "f32" (default, exact) or "bf16" (bf16 MFMA with
 * fp32 accumulation — the MI355X-native mixed-precision mode; replaces
 * the reference's NetParameter default_forward_type/default_backward_type
 * FLOAT16 machinery, net.cpp:100-156 / type.hpp:13-47, with fp32 storage
 * and tensor-core math instead of fp16 storage). */
int caffe_set_compute(const char* dtype);
/* rank/world for the sharded data feed without a communicator (tests);
 * caffe_comm_init sets the same engine fields. */
int caffe_set_rank_world(int rank, int world);
/* data-stream position (LMDB cursor analog; Solver::Step drives it). */
int caffe_set_data_iter(uint64_t iter);
/* multi-node ncclUniqueId bootstrap over TCP (the reference's
 * Clusters/MPI_Bcast replacement, clusters.cpp + parallel.cpp:42-45):
 * global rank 0 serves the 128-byte id to world-1 clients; the `caffe`
 * CLI drives this from CAFFE_NNODES/CAFFE_NODE_RANK/MASTER_ADDR/PORT. */
int caffe_uid_serve(const uint8_t* uid, int port, int nclients);
int caffe_uid_fetch(uint8_t* out, const char* host, int port,
                    int timeout_s);
/* Caffe::set_random_seed (common.cpp); rank offsets seed like
 * parallel.cpp:179-187 */
int caffe_set_random_seed(uint64_t seed);
/* synthetic data source configuration (replaces the LMDB DataReader,
 * data_reader.cpp, for dataset-less benching): shape of one sample and the
 * label range */
int caffe_set_synthetic_shape(int c, int h, int w, int num_classes);
/* enable per-kernel-class event timing (for roofline measurement) */
int caffe_set_perf_timing(int enable);

/* SolverRegistry::CreateSolver + Solver::Solve entry
 * (tools/caffe.cpp:213-241, solver.cpp:187).  batch_override > 0 replaces
 * the train data layer's batch_size (the reference divides the prototxt
 * batch across GPUs, parallel.cpp:284-348). */
caffe_solver_t caffe_solver_create(const char* solver_prototxt_path,
                                   int batch_override);
caffe_solver_t caffe_solver_create_from_text(const char* solver_prototxt,
                                             int batch_override);
void caffe_solver_free(caffe_solver_t s);
/* Solver::Step(iters) (solver.cpp:187-353): forward, backward with the
 * bucketed all-reduce overlapped on the side stream, fused SGD update */
int caffe_solver_step(caffe_solver_t s, int iters);
long caffe_solver_iter(caffe_solver_t s);
/* current net loss (the reference additionally smooths over the display
 * window, solver.cpp:606-617; this returns the raw last-iteration loss)
 * — syncs */
float caffe_solver_loss(caffe_solver_t s);
caffe_net_t caffe_solver_net(caffe_solver_t s);

/* Solver::Snapshot / Restore (solver.cpp:542-604; .caffemodel binaryproto
 * weights + .solverstate with SGD history, sgd_solver.cpp:262-353) */
int caffe_solver_snapshot(caffe_solver_t s);
int caffe_solver_restore(caffe_solver_t s, const char* state_path);
/* Net weight interop (net.cpp:1055-1248) */
int caffe_net_save_weights(caffe_net_t n, const char* path);
int caffe_net_load_weights(caffe_net_t n, const char* path);

/* one-process-per-GPU collective bootstrap (replaces Clusters::Init +
 * MPI_Bcast of ncclUniqueId, clusters.cpp:8 / parallel.cpp:42-45):
 * rank 0 generates the 128-byte id, the launcher distributes it, every
 * rank calls comm_init */
int caffe_comm_unique_id(uint8_t out[128]);
/* world-1 RCCL linkage/call smoke (GPU mode) */
int caffe_comm_selftest(void);
int caffe_comm_init(caffe_solver_t s, int rank, int world,
                    const uint8_t id[128]);
/* initial weight broadcast from rank 0 (P2PSync::on_start,
 * parallel.cpp:208-227) */
int caffe_comm_bcast_weights(caffe_solver_t s);
/* CPU-mode collective for multi-process gloo tests: the callback receives
 * (grad_ptr, count, userdata) at every bucket flush */
typedef void (*caffe_allreduce_cb)(float*, long, void*);
int caffe_comm_set_callback(caffe_solver_t s, caffe_allreduce_cb cb,
                            void* ud, int world);

/* Net construction for inference/parity runs (Net::Init, net.cpp:64) */
caffe_net_t caffe_net_create(const char* net_prototxt_path, int phase,
                             int batch_override);
void caffe_net_free(caffe_net_t n);
int caffe_net_forward(caffe_net_t n);           /* net.cpp:692 */
int caffe_net_backward(caffe_net_t n);          /* net.cpp:1031 */
float caffe_net_loss(caffe_net_t n);            /* syncs */

/* blob access by name (Net::blob_by_name, net.hpp): data or diff */
int caffe_net_blob_shape(caffe_net_t n, const char* name, int* shape_out,
                         int max_dims, int* ndims_out);
int caffe_net_blob_get(caffe_net_t n, const char* name, int diff,
                       float* out, long count);
int caffe_net_blob_set(caffe_net_t n, const char* name, int diff,
                       const float* in, long count);

/* learnable params (Net::learnable_params, net.cpp:1350): idx ordering is
 * the arena (backward-completion) order */
/* pycaffe-shim surface: zero-copy CPU pointers with SyncedMemory head
 * semantics (mutable access dirties the host copy; next GPU use re-syncs
 * — the contract python/caffe/_caffe.cpp exposed through Blob.data /
 * Blob.diff), plus layer enumeration for net.blobs/net.params/net.layers
 * dict shapes. */
float* caffe_net_blob_cpu_ptr(caffe_net_t n, const char* name, int diff,
                              int writable);
int caffe_net_num_layers(caffe_net_t n);
int caffe_net_blob_names(caffe_net_t n, char* out, int cap);
int caffe_net_layer_info(caffe_net_t n, int idx, char* name_out,
                         int name_cap, char* type_out, int type_cap,
                         int* num_blobs_out);
int caffe_net_layer_blob_shape(caffe_net_t n, const char* lname, int bidx,
                               int* shape_out, int max_dims,
                               int* ndims_out);
float* caffe_net_layer_blob_cpu_ptr(caffe_net_t n, const char* lname,
                                    int bidx, int diff, int writable);

int caffe_net_num_params(caffe_net_t n);
int caffe_net_param_info(caffe_net_t n, int idx, char* layer_name_out,
                         int name_cap, int* blob_idx_out, long* count_out);
int caffe_net_param_get(caffe_net_t n, int idx, int diff, float* out,
                        long count);
int caffe_net_param_set(caffe_net_t n, int idx, const float* in, long count);

/* device + perf introspection */
int caffe_device_synchronize(void);
/* perf counters per kernel class: returns number of classes; each row of
 * `names` gets a \0-terminated class name (cap bytes), and launches /
 * flops / bytes / ns land in the parallel arrays */
int caffe_perf_snapshot(char* names, int name_cap, long* launches,
                        double* flops, double* bytes, double* ns, int max_rows);
int caffe_perf_reset(void);

#ifdef __cplusplus
}
#endif
#endif /* CAFFE_AMD_H_ */
