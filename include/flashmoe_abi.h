/* C-ABI boundary of the MI355X-native FlashDMoE hot path.
 *
 * Each entry point cites the reference interface it replaces
 * (file:line into /root/reference). No torch types: plain pointers,
 * sizes and an opaque HIP stream. All device pointers are raw HIP
 * device memory on the current device; the caller owns input/output
 * buffers, the library owns its internal workspace.
 *
 * Threading model: one process per GPU, all calls from
 * one host thread per process, kernels enqueued on the caller's stream.
 *
 * The Python mirror (flashmoe_amd/moe.py, ctypes) reproduces the
 * reference's pybind11 surface `flashmoe._C`
 * (csrc/python_bindings.cu:194-217) on top of these functions; the
 * binding stub a maintainer would add upstream is in INTEGRATION.md.
 */
#ifndef FLASHMOE_ABI_H
#define FLASHMOE_ABI_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* Error codes (negative) / 0 on success. fm_last_error() returns a
 * human-readable message for the most recent failure on this thread. */
#define FM_OK 0
#define FM_ERR_HIP (-1)          /* HIP runtime failure */
#define FM_ERR_STATE (-2)        /* initialize/finalize ordering */
#define FM_ERR_SHAPE (-3)        /* shape mismatch vs frozen config */
#define FM_ERR_UNSUPPORTED (-4)  /* dtype/feature not built */

/* Mirror of csrc/flashmoe_config.json (schema:
 * csrc/flashmoe_config.schema.json). dtype: 0 fp32, 1 tf32(=fp32 on
 * CDNA4 - no xf32), 2 bf16, 3 fp16, 4 fp8e4m3 expert weights with bf16
 * activations (W8A16), 5 MX-block-scaled fp8: fp8e4m3 weights AND
 * runtime-quantized fp8 activations (per-64-element E8M0 block scales)
 * on the CDNA4 scaled MFMA; weights are passed exactly as for dtype 4,
 * activation quantization is internal. dtype 5 requires H, P multiples
 * of 128. */
typedef struct fm_config {
  int32_t num_experts;
  int32_t expert_top_k;
  int32_t capacity_factor;
  int32_t drop_tokens;
  int32_t hidden_act; /* 0 relu, 1 gelu */
  int32_t hidden_size;        /* H */
  int32_t intermediate_size;  /* P */
  int32_t sequence_len;  /* sequence_len * mini_batch (= S, the token
                          * count) must be a positive multiple of 128:
                          * the gate tiles tokens in blocks of 128 */
  int32_t mini_batch;
  int32_t dtype;
  int32_t is_training;
} fm_config;

/* Replaces flashmoe::initialize (python_bindings.cu:157-159,
 * bootstrap.cuh:533-547): freezes the config, selects the device,
 * allocates the library workspace (tokenIds, eC, xM, O32) sized from the
 * config. rank/world describe the EP layout (nLx = num_experts/world,
 * uniform split, bootstrap.cuh:36-52); comms bootstrap itself lives on
 * the Python side (torch.distributed over RCCL). */
int fm_initialize(const fm_config* cfg, int rank, int world_size);

/* Replaces flashmoe::finalize (python_bindings.cu:163-168,
 * bootstrap.cuh:561-588): frees the workspace. */
int fm_finalize(void);

/* Replaces _C.get_compiled_config (python_bindings.cu:170-183 /
 * flashmoe/ops.py:63-71): S, H, E, P, PX, element byte size. */
int fm_get_compiled_config(int64_t* S, int64_t* H, int64_t* E, int64_t* P,
                           int64_t* PX, int64_t* element_size);

/* Replaces _C.get_num_local_experts (python_bindings.cu:185-189). */
int fm_get_num_local_experts(void);

/* THE hot path. Replaces _C.moe_forward (python_bindings.cu:17-151) for
 * the single-rank case: gate -> route -> expert FFN -> combine on this
 * rank's tokens against this rank's nLx experts (world 1: all E).
 *
 * stream:    hipStream_t as void* (0 = default stream)
 * x:         [S, H] Element, device, contiguous
 * gate_w:    flat E*H Element buffer (the torch [H,E] tensor's storage,
 *            used as a row-major [E,H] matrix - reference quirk,
 *            moe.cuh:107-109; see oracle/moe_oracle.py docstring)
 * expert_w:  [nLx, 2, P, H] Element ([e][0]=Wup [P,H]; [e][1] flat,
 *            used as [H,P] - moe.cuh:114-116)
 * b_up/b_dn: [nLx, P] / [nLx, H] Element or NULL (reference zero-fills,
 *            python_bindings.cu:80-82)
 * gate_out:  [S, PX] Element out (softmax probs, moe.cuh:128-131)
 * moe_out:   [S, H] Element out
 * S:         token count this call (validated against frozen config)
 *
 * Asynchronous: enqueues on `stream`; no sync, no allocation. */
int fm_moe_forward(void* stream, const void* x, const void* gate_w,
                   const void* expert_w, const void* b_up, const void* b_dn,
                   void* gate_out, void* moe_out, int64_t S);

/* --- Staged entry points for the expert-parallel (multi-GPU) path.
 * The reference runs these stages inside one kernel with one-sided
 * NVSHMEM/P2P exchange between dispatch and FFN (moe.cuh:134-143); here
 * the host pipeline calls them around the RCCL all-to-all. --- */

/* Gate only: writes gate_out, and the library's tokenIds/eC workspace
 * (gate.cuh:474-720 semantics). */
int fm_gate_forward(void* stream, const void* x, const void* gate_w,
                    void* gate_out, int64_t S);

/* Copy routing results out: routed[e] = min(eC[e], EC) and the ordered
 * token slots (token index + probSum) per expert, for the host to build
 * the all-to-all. Device-to-host; synchronizes `stream`. */
int fm_read_routing(void* stream, uint32_t* routed_counts /* [E] */,
                    uint32_t* token_idx /* [E*EC] */,
                    float* prob_sum /* [E*EC] */);

/* Expert FFN on pre-packed rows (identity gather): rows [n_rows, H]
 * belong to local expert `local_e`; writes z = (act(rows@WupT+b))@WdnT+b
 * to out_rows [n_rows, H] (processor.cuh:685-751). */
int fm_expert_ffn(void* stream, const void* rows, const void* expert_w,
                  const void* b_up, const void* b_dn, void* out_rows,
                  int64_t n_rows, int32_t local_e);

/* --- Capacity-padded EP pipeline (the product multi-GPU path). The
 * exchange unit is the reference's symmetric-heap cell layout
 * (types.cuh:1014-1032): a fixed [E, EC, H] buffer, expert-major,
 * capacity-padded - so the all_to_all has STATIC equal splits and the
 * step needs no host synchronization at all. Rows past each expert's
 * routed count carry garbage and are dropped at the source. --- */

/* Gather x rows into the [E, EC, H] dispatch buffer per the last
 * fm_gate_forward's routing (os/packet.cuh:20-286 dispatch semantics). */
int fm_pack_dispatch(void* stream, const void* x, void* sendbuf);

/* FFN over the received padded buffer: rows = [n_segs, EC, H] with
 * seg_expert_dev = device int32[n_segs] mapping segment -> local expert
 * (canonical order: source-rank major). Two kernel launches total. */
int fm_expert_ffn_segments(void* stream, const void* rows,
                           const void* seg_expert_dev, int32_t n_segs,
                           const void* expert_w, void* out_rows);

/* Combine the returned [E, EC, H] buffer at the source using the local
 * routing metadata (scale = gate_out/probSum; k==1 unscaled) and write
 * moe_out (processor.cuh:44-205). */
int fm_combine_padded(void* stream, const void* returned,
                      const void* gate_out, void* moe_out, int64_t S);

/* Batched fm_expert_ffn over all local experts: rows are packed
 * expert-major; offsets = host array of n_experts+1 row offsets
 * (offsets[le]..offsets[le+1] belong to local expert le). One call per
 * EP step instead of nLx ctypes crossings. */
int fm_expert_ffn_grouped(void* stream, const void* rows,
                          const int64_t* offsets, int32_t n_experts,
                          const void* expert_w, const void* b_up,
                          const void* b_dn, void* out_rows);

/* Combine pre-scaled return rows into moe_out: for i < n_rows,
 * moe_out[token_idx[i]] += scale[i] * rows[i] (k>1 path,
 * processor.cuh:126-168); k==1: unscaled overwrite. Caller passes
 * scale[i] = gate_out[t,e]/probSum[t]. zero_first resets the fp32
 * accumulator before adding. finalize converts the accumulator into
 * moe_out. */
int fm_combine(void* stream, const void* rows, const uint32_t* token_idx,
               const float* scale, int64_t n_rows, int32_t zero_first);
int fm_combine_finalize(void* stream, void* moe_out, int64_t S);

/* Phase-timed variant of fm_moe_forward for roofline measurement
 * (replaces the reference's in-API benchmark loop semantics,
 * moe.cuh:146-184 forwardHostBench): HIP events are recorded between the
 * phases on `stream`, the stream is synchronized, and ms[0..3] receives
 * {gate, expert-up GEMM, expert-down GEMM+combine, other (memset/cast)}
 * durations in milliseconds. Not for use inside a timed region. */
int fm_moe_forward_phased(void* stream, const void* x, const void* gate_w,
                          const void* expert_w, const void* b_up,
                          const void* b_dn, void* gate_out, void* moe_out,
                          int64_t S, float ms[4]);

/* GPU-resident routing export for the EP pipeline: routed_dev = u32[E]
 * clipped counts (min(eC, EC)), tps_dev = u32[E][EC][2] (tokenIdx,
 * probSum bits). Device pointers; asynchronous (no host sync) - this is
 * what keeps the multi-GPU dispatch planning off the host critical path
 * (replaces the reference's in-kernel decode, os/packet.cuh:288-455). */
int fm_export_routing(void* stream, void* routed_dev, void* tps_dev);

/* --- Opt-in one-sided P2P transport for the EP exchange (the
 * reference's intra-node mode: direct stores into hipIpc-mapped peer
 * heap cells + system-scope 8-byte seq-tagged signals,
 * os/packet.cuh:214-258; heap cell layout = the padded pipeline's).
 * Enable from Python with FLASHMOE_P2P=1; the all_to_all path stays the
 * default until multi-GPU validation. --- */
int fm_heap_init(void);
int fm_heap_handle(void* out64 /* hipIpcMemHandle_t, 64 B */);
int fm_heap_connect(const void* handles /* world x 64 B; NULL = self only */);
int fm_heap_ptrs(void** recv /* [world,nLx,EC,H] */, void** ret /* [E,EC,H] */);
/* write routed rows into owners' heaps + signal; then wait for all
 * sources' cells for MY experts (bounded spin; prints on timeout) */
int fm_dispatch_p2p(void* stream, const void* x);
/* write FFN results back into sources' return heaps + signal; wait for
 * my own return cells */
int fm_return_p2p(void* stream, const void* ffn_out);
/* A timed-out in-kernel flag wait (bounded spin gave up) poisons the
 * exchange: the kernel sets a host-mapped error word, the NEXT
 * fm_dispatch_p2p/fm_return_p2p call returns FM_ERR_HIP without
 * launching, and this call checks definitively (synchronizes the
 * stream, returns FM_ERR_HIP if the word is set, then clears it so the
 * caller may retry). */
int fm_p2p_error_check(void* stream);

/* Training-mode auxiliary-loss accumulators (gate.cuh:273-299,763-773;
 * types.cuh:936-958): gML[e] = mean softmax prob of expert e over the
 * last forward's tokens, gMeC[e] = fraction routed to e (pre-capacity).
 * Host arrays of E floats; synchronizes the stream. is_training=1 only. */
int fm_read_aux_loss(void* stream, float* gML, float* gMeC);

/* Version/introspection */
const char* fm_last_error(void);
int fm_built_for_gfx950(void);

#ifdef __cplusplus
}
#endif
#endif /* FLASHMOE_ABI_H */
