/* C API smoke: import a GraphDef, feed x, fetch y = x*w; then load the
 * SavedModel in dir argv[2] and run its serving signature. */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include "c_api/c_api.h"

static char* read_file(const char* path, size_t* len) {
  FILE* f = fopen(path, "rb");
  if (!f) return NULL;
  fseek(f, 0, SEEK_END);
  *len = ftell(f);
  fseek(f, 0, SEEK_SET);
  char* buf = malloc(*len);
  if (fread(buf, 1, *len, f) != *len) { fclose(f); free(buf); return NULL; }
  fclose(f);
  return buf;
}

#define CHECK(cond, msg) do { if (!(cond)) { \
  fprintf(stderr, "FAIL: %s\n", msg); return 1; } } while (0)

int main(int argc, char** argv) {
  CHECK(argc >= 3, "usage: c_api_smoke graph.pb savedmodel_dir");
  size_t len;
  char* gd = read_file(argv[1], &len);
  CHECK(gd != NULL, "read graph.pb");

  TF_Status* st = TF_NewStatus();
  TF_Graph* graph = TF_NewGraph();
  TF_Buffer* buf = TF_NewBufferFromString(gd, len);
  free(gd);
  TF_ImportGraphDefOptions* iopts = TF_NewImportGraphDefOptions();
  TF_GraphImportGraphDef(graph, buf, iopts, st);
  CHECK(TF_GetCode(st) == TF_OK, TF_Message(st));
  TF_DeleteBuffer(buf);
  TF_DeleteImportGraphDefOptions(iopts);

  TF_Operation* x_op = TF_GraphOperationByName(graph, "x");
  TF_Operation* y_op = TF_GraphOperationByName(graph, "y");
  CHECK(x_op && y_op, "ops present");
  CHECK(strcmp(TF_OperationOpType(x_op), "Placeholder") == 0, "x type");

  TF_SessionOptions* sopts = TF_NewSessionOptions();
  TF_Session* sess = TF_NewSession(graph, sopts, st);
  CHECK(TF_GetCode(st) == TF_OK, TF_Message(st));

  int64_t dims[2] = {1, 2};
  TF_Tensor* x = TF_AllocateTensor(TF_FLOAT, dims, 2, 8);
  ((float*)TF_TensorData(x))[0] = 1.0f;
  ((float*)TF_TensorData(x))[1] = 2.0f;
  TF_Output in = {x_op, 0}, out = {y_op, 0};
  TF_Tensor* y = NULL;
  TF_SessionRun(sess, NULL, &in, &x, 1, &out, &y, 1, NULL, 0, NULL, st);
  CHECK(TF_GetCode(st) == TF_OK, TF_Message(st));
  CHECK(y != NULL && TF_NumDims(y) == 2, "y shape");
  float* yv = (float*)TF_TensorData(y);
  /* w = [[3],[4]] -> y = 1*3+2*4 = 11 */
  CHECK(yv[0] == 11.0f, "y value");
  TF_DeleteTensor(x);
  TF_DeleteTensor(y);
  TF_DeleteSession(sess, st);
  TF_DeleteGraph(graph);

  /* ---- SavedModel ---- */
  TF_Graph* g2 = TF_NewGraph();
  const char* tags[1] = {"serve"};
  TF_Buffer* mgd = TF_NewBuffer();
  TF_Session* s2 = TF_LoadSessionFromSavedModel(sopts, NULL, argv[2], tags, 1,
                                                g2, mgd, st);
  CHECK(TF_GetCode(st) == TF_OK, TF_Message(st));
  CHECK(mgd->length > 0, "meta graph bytes");
  TF_Operation* x2 = TF_GraphOperationByName(g2, "x");
  TF_Operation* y2 = TF_GraphOperationByName(g2, "y");
  CHECK(x2 && y2, "saved model ops");
  TF_Tensor* xv2 = TF_AllocateTensor(TF_FLOAT, dims, 2, 8);
  ((float*)TF_TensorData(xv2))[0] = 2.0f;
  ((float*)TF_TensorData(xv2))[1] = 0.5f;
  TF_Output in2 = {x2, 0}, out2 = {y2, 0};
  TF_Tensor* yv2 = NULL;
  TF_SessionRun(s2, NULL, &in2, &xv2, 1, &out2, &yv2, 1, NULL, 0, NULL, st);
  CHECK(TF_GetCode(st) == TF_OK, TF_Message(st));
  /* restored w = [[3],[4]] -> 2*3+0.5*4 = 8 */
  CHECK(((float*)TF_TensorData(yv2))[0] == 8.0f, "saved model value");

  printf("C_API_OK\n");
  return 0;
}
