/* C predict API for the anand-mxnet-amd framework (reference parity:
 * include/mxnet/c_predict_api.h).  Link against the extension .so
 * (anand_mxnet_amd/_C.cpython-*.so) and libpython; call from any C/C++
 * program to run inference on exported `-symbol.json` + `.params`
 * checkpoints.  All functions return 0 on success, -1 on failure with
 * the message available from MXAGetLastError(). */
#ifndef MXA_PREDICT_H_
#define MXA_PREDICT_H_

#ifdef __cplusplus
extern "C" {
#endif

typedef void* MXAPredictorHandle;

const char* MXAGetLastError(void);

/* symbol_json: file path OR the JSON text itself; dev_type 1=cpu 2=gpu */
int MXAPredCreate(const char* symbol_json, const char* params_file,
                  int dev_type, int dev_id, unsigned num_inputs,
                  const char** input_names, MXAPredictorHandle* out);
int MXAPredSetInput(MXAPredictorHandle h, const char* name,
                    const float* data, const long* shape, unsigned ndim);
int MXAPredForward(MXAPredictorHandle h);
int MXAPredGetOutputShape(MXAPredictorHandle h, unsigned index, long* shape,
                          unsigned* ndim);
int MXAPredGetOutput(MXAPredictorHandle h, unsigned index, float* out,
                     long size);
int MXAPredFree(MXAPredictorHandle h);

#ifdef __cplusplus
}
#endif
#endif  /* MXA_PREDICT_H_ */
