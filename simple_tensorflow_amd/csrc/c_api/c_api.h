/* C API for the MI355X-native runtime — stable-ABI subset of the reference's
 * c/c_api.h (TF_NewSession:939, TF_SessionRun:999, TF_GraphImportGraphDef:849,
 * TF_LoadSessionFromSavedModel:956). Same call shapes and semantics; enums
 * match the reference numerically (TF_Code == error::Code, TF_DataType ==
 * types.proto DataType). */
#ifndef STF_C_API_H_
#define STF_C_API_H_

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef enum TF_Code {
  TF_OK = 0,
  TF_CANCELLED = 1,
  TF_UNKNOWN = 2,
  TF_INVALID_ARGUMENT = 3,
  TF_DEADLINE_EXCEEDED = 4,
  TF_NOT_FOUND = 5,
  TF_ALREADY_EXISTS = 6,
  TF_PERMISSION_DENIED = 7,
  TF_UNAUTHENTICATED = 16,
  TF_RESOURCE_EXHAUSTED = 8,
  TF_FAILED_PRECONDITION = 9,
  TF_ABORTED = 10,
  TF_OUT_OF_RANGE = 11,
  TF_UNIMPLEMENTED = 12,
  TF_INTERNAL = 13,
  TF_UNAVAILABLE = 14,
  TF_DATA_LOSS = 15,
} TF_Code;

typedef enum TF_DataType {
  TF_FLOAT = 1,
  TF_DOUBLE = 2,
  TF_INT32 = 3,
  TF_UINT8 = 4,
  TF_INT16 = 5,
  TF_INT8 = 6,
  TF_STRING = 7,
  TF_INT64 = 9,
  TF_BOOL = 10,
  TF_BFLOAT16 = 14,
  TF_HALF = 19,
} TF_DataType;

typedef struct TF_Status TF_Status;
typedef struct TF_Buffer {
  const void* data;
  size_t length;
  void (*data_deallocator)(void* data, size_t length);
} TF_Buffer;
typedef struct TF_Tensor TF_Tensor;
typedef struct TF_SessionOptions TF_SessionOptions;
typedef struct TF_Graph TF_Graph;
typedef struct TF_Operation TF_Operation;
typedef struct TF_Session TF_Session;
typedef struct TF_ImportGraphDefOptions TF_ImportGraphDefOptions;

typedef struct TF_Output {
  TF_Operation* oper;
  int index;
} TF_Output;

/* ---- status ---- */
TF_Status* TF_NewStatus(void);
void TF_DeleteStatus(TF_Status*);
void TF_SetStatus(TF_Status*, TF_Code code, const char* msg);
TF_Code TF_GetCode(const TF_Status*);
const char* TF_Message(const TF_Status*);

/* ---- buffer ---- */
TF_Buffer* TF_NewBuffer(void);
TF_Buffer* TF_NewBufferFromString(const void* proto, size_t len);
void TF_DeleteBuffer(TF_Buffer*);

/* ---- tensor ---- */
TF_Tensor* TF_AllocateTensor(TF_DataType, const int64_t* dims, int num_dims,
                             size_t len);
TF_Tensor* TF_NewTensor(TF_DataType, const int64_t* dims, int num_dims,
                        void* data, size_t len,
                        void (*deallocator)(void* data, size_t len, void* arg),
                        void* deallocator_arg);
void TF_DeleteTensor(TF_Tensor*);
TF_DataType TF_TensorType(const TF_Tensor*);
int TF_NumDims(const TF_Tensor*);
int64_t TF_Dim(const TF_Tensor*, int dim_index);
size_t TF_TensorByteSize(const TF_Tensor*);
void* TF_TensorData(const TF_Tensor*);

/* ---- graph ---- */
TF_Graph* TF_NewGraph(void);
void TF_DeleteGraph(TF_Graph*);
TF_ImportGraphDefOptions* TF_NewImportGraphDefOptions(void);
void TF_DeleteImportGraphDefOptions(TF_ImportGraphDefOptions*);
void TF_ImportGraphDefOptionsSetPrefix(TF_ImportGraphDefOptions*,
                                       const char* prefix);
void TF_GraphImportGraphDef(TF_Graph* graph, const TF_Buffer* graph_def,
                            const TF_ImportGraphDefOptions* options,
                            TF_Status* status);
void TF_GraphToGraphDef(TF_Graph* graph, TF_Buffer* output_graph_def,
                        TF_Status* status);
TF_Operation* TF_GraphOperationByName(TF_Graph* graph, const char* oper_name);
const char* TF_OperationName(TF_Operation* oper);
const char* TF_OperationOpType(TF_Operation* oper);

/* ---- session ---- */
TF_SessionOptions* TF_NewSessionOptions(void);
void TF_DeleteSessionOptions(TF_SessionOptions*);
TF_Session* TF_NewSession(TF_Graph* graph, const TF_SessionOptions* opts,
                          TF_Status* status);
void TF_CloseSession(TF_Session*, TF_Status*);
void TF_DeleteSession(TF_Session*, TF_Status*);
void TF_SessionRun(TF_Session* session, const TF_Buffer* run_options,
                   const TF_Output* inputs, TF_Tensor* const* input_values,
                   int ninputs, const TF_Output* outputs,
                   TF_Tensor** output_values, int noutputs,
                   const TF_Operation* const* target_opers, int ntargets,
                   TF_Buffer* run_metadata, TF_Status* status);
TF_Session* TF_LoadSessionFromSavedModel(
    const TF_SessionOptions* session_options, const TF_Buffer* run_options,
    const char* export_dir, const char* const* tags, int tags_len,
    TF_Graph* graph, TF_Buffer* meta_graph_def, TF_Status* status);

const char* TF_Version(void);

#ifdef __cplusplus
} /* extern "C" */
#endif
#endif /* STF_C_API_H_ */
