/* C ABI for embedding the participant in mobile/desktop apps.
 *
 * API parity with the reference's xaynet-mobile FFI
 * (rust/xaynet-mobile/src/ffi/participant.rs:27-470, ffi/settings.rs:46-274,
 * cbindgen header): settings builder, key generation, participant
 * new/tick/save/restore/set_model/global_model/destroy, tick flag bitmask.
 *
 * Build: libxaynet_ffi.so (build_ffi.py). All functions are thread-compatible
 * but a single participant must not be used from two threads concurrently.
 */
#ifndef XAYNET_FFI_H
#define XAYNET_FFI_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- return codes ---- */
#define XAYNET_FFI_OK 0
#define XAYNET_FFI_ERR_NULLPTR (-1)
#define XAYNET_FFI_ERR_INVALID (-2)
#define XAYNET_FFI_GLOBALMODEL_NONE 1
#define XAYNET_FFI_ERR_GLOBALMODEL_LEN (-3)
#define XAYNET_FFI_ERR_GLOBALMODEL_DATATYPE (-4)
#define XAYNET_FFI_ERR_GLOBALMODEL_IO (-5)
#define XAYNET_FFI_ERR_SETMODEL_DATATYPE (-6)
#define XAYNET_FFI_ERR_SETMODEL_LEN (-7)
#define XAYNET_FFI_ERR_RESTORE (-8)

/* ---- tick flags (reference participant.rs:61-73) ---- */
#define XAYNET_FFI_PARTICIPANT_TASK_NONE (1 << 0)
#define XAYNET_FFI_PARTICIPANT_TASK_SUM (1 << 1)
#define XAYNET_FFI_PARTICIPANT_TASK_UPDATE (1 << 2)
#define XAYNET_FFI_PARTICIPANT_SHOULD_SET_MODEL (1 << 3)
#define XAYNET_FFI_PARTICIPANT_MADE_PROGRESS (1 << 4)
#define XAYNET_FFI_PARTICIPANT_NEW_GLOBALMODEL (1 << 5)

/* ---- model data types (mask::DataType) ---- */
#define XAYNET_FFI_DATATYPE_F32 0
#define XAYNET_FFI_DATATYPE_F64 1
#define XAYNET_FFI_DATATYPE_I32 2
#define XAYNET_FFI_DATATYPE_I64 3

typedef struct XaynetFfiSettings XaynetFfiSettings;
typedef struct XaynetFfiParticipant XaynetFfiParticipant;

typedef struct {
    uint8_t secret[32]; /* Ed25519 seed */
    uint8_t public_[32];
} XaynetFfiKeyPair;

typedef struct {
    uint8_t* data;
    size_t len;
} XaynetFfiByteBuffer;

/* ---- settings ---- */
XaynetFfiSettings* xaynet_ffi_settings_new(void);
int xaynet_ffi_settings_destroy(XaynetFfiSettings* s);
int xaynet_ffi_settings_set_url(XaynetFfiSettings* s, const char* url);
int xaynet_ffi_settings_set_scalar(XaynetFfiSettings* s, double scalar);
int xaynet_ffi_settings_set_keys(XaynetFfiSettings* s, const XaynetFfiKeyPair* keys);
int xaynet_ffi_check_settings(const XaynetFfiSettings* s);

const XaynetFfiKeyPair* xaynet_ffi_generate_key_pair(void);
int xaynet_ffi_forget_key_pair(const XaynetFfiKeyPair* kp);

/* ---- participant ---- */
XaynetFfiParticipant* xaynet_ffi_participant_new(const XaynetFfiSettings* s);
int xaynet_ffi_participant_destroy(XaynetFfiParticipant* p);

/* one state-machine transition; returns the flag bitmask or ERR_NULLPTR */
int xaynet_ffi_participant_tick(XaynetFfiParticipant* p);

int xaynet_ffi_participant_set_model(XaynetFfiParticipant* p, const void* buffer,
                                     unsigned char data_type, unsigned int len);
/* copies the current global model into buffer (len elements of data_type);
 * returns OK / GLOBALMODEL_NONE / errors */
int xaynet_ffi_participant_global_model(XaynetFfiParticipant* p, void* buffer,
                                        unsigned char data_type, unsigned int len);
/* current round model schema; returns OK and fills outputs (len 0 / type -1
 * before the first round-params fetch) */
int xaynet_ffi_participant_local_model_config(const XaynetFfiParticipant* p,
                                              int* data_type_out, uint64_t* len_out);

/* serialize + consume (participant must still be destroyed) */
XaynetFfiByteBuffer* xaynet_ffi_participant_save(XaynetFfiParticipant* p);
// xaynet-mobile-compatible checkpoint (bincode SerializableState); the
// generic restore auto-detects both formats
XaynetFfiByteBuffer* xaynet_ffi_participant_save_reference(XaynetFfiParticipant* p);
XaynetFfiParticipant* xaynet_ffi_participant_restore(const char* url,
                                                     const XaynetFfiByteBuffer* state);
int xaynet_ffi_byte_buffer_destroy(XaynetFfiByteBuffer* b);

#ifdef __cplusplus
}
#endif

#endif /* XAYNET_FFI_H */
