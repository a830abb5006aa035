/*!
 * migbm C API — flat C ABI compatible with the reference's include/LightGBM/c_api.h
 * (function names, signatures and semantics match for every implemented entry point so
 * existing LGBM_* clients can link against lib_migbm). Fresh implementation.
 */
#ifndef MIGBM_C_API_H_
#define MIGBM_C_API_H_

#include <cstdint>
#include <cstdlib>

#ifdef __cplusplus
#define MIGBM_EXTERN_C extern "C"
#else
#define MIGBM_EXTERN_C
#endif

#define LIGHTGBM_C_EXPORT MIGBM_EXTERN_C __attribute__((visibility("default")))

typedef void* DatasetHandle;
typedef void* BoosterHandle;

#define C_API_DTYPE_FLOAT32 (0)
#define C_API_DTYPE_FLOAT64 (1)
#define C_API_DTYPE_INT32 (2)
#define C_API_DTYPE_INT64 (3)

#define C_API_PREDICT_NORMAL (0)
#define C_API_PREDICT_RAW_SCORE (1)
#define C_API_PREDICT_LEAF_INDEX (2)
#define C_API_PREDICT_CONTRIB (3)

#define C_API_MATRIX_TYPE_CSR (0)
#define C_API_MATRIX_TYPE_CSC (1)

#define C_API_FEATURE_IMPORTANCE_SPLIT (0)
#define C_API_FEATURE_IMPORTANCE_GAIN (1)

LIGHTGBM_C_EXPORT const char* LGBM_GetLastError();
LIGHTGBM_C_EXPORT int LGBM_RegisterLogCallback(void (*callback)(const char*));
LIGHTGBM_C_EXPORT int LGBM_SetMaxThreads(int num_threads);
LIGHTGBM_C_EXPORT int LGBM_GetMaxThreads(int* out);
LIGHTGBM_C_EXPORT int LGBM_SetLastError(const char* msg);
LIGHTGBM_C_EXPORT int LGBM_DumpParamAliases(int64_t buffer_len, int64_t* out_len,
                                            char* out_str);
LIGHTGBM_C_EXPORT int LGBM_GetSampleCount(int32_t num_total_row, const char* parameters,
                                          int* out);
LIGHTGBM_C_EXPORT int LGBM_SampleIndices(int32_t num_total_row, const char* parameters,
                                         void* out, int32_t* out_len);

// ---- Dataset
LIGHTGBM_C_EXPORT int LGBM_DatasetCreateFromFile(const char* filename, const char* parameters,
                                                 const DatasetHandle reference,
                                                 DatasetHandle* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetCreateFromMat(const void* data, int data_type, int32_t nrow,
                                                int32_t ncol, int is_row_major,
                                                const char* parameters,
                                                const DatasetHandle reference,
                                                DatasetHandle* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetCreateFromMats(int32_t nmat, const void** data, int data_type,
                                                 int32_t* nrow, int32_t ncol, int is_row_major,
                                                 const char* parameters,
                                                 const DatasetHandle reference,
                                                 DatasetHandle* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetCreateFromCSR(const void* indptr, int indptr_type,
                                                const int32_t* indices, const void* data,
                                                int data_type, int64_t nindptr, int64_t nelem,
                                                int64_t num_col, const char* parameters,
                                                const DatasetHandle reference,
                                                DatasetHandle* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetCreateFromCSC(const void* col_ptr, int col_ptr_type,
                                                const int32_t* indices, const void* data,
                                                int data_type, int64_t ncol_ptr, int64_t nelem,
                                                int64_t num_row, const char* parameters,
                                                const DatasetHandle reference,
                                                DatasetHandle* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetCreateFromSampledColumn(double** sample_data,
                                                          int** sample_indices,
                                                          int32_t ncol,
                                                          const int* num_per_col,
                                                          int32_t num_sample_row,
                                                          int32_t num_local_row,
                                                          int64_t num_dist_row,
                                                          const char* parameters,
                                                          DatasetHandle* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetCreateByReference(const DatasetHandle reference,
                                                    int64_t num_total_row,
                                                    DatasetHandle* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetInitStreaming(DatasetHandle dataset, int32_t has_weights,
                                                int32_t has_init_scores,
                                                int32_t has_queries, int32_t nclasses,
                                                int32_t nthreads, int32_t omp_max_threads);
LIGHTGBM_C_EXPORT int LGBM_DatasetPushRows(DatasetHandle dataset, const void* data,
                                           int data_type, int32_t nrow, int32_t ncol,
                                           int32_t start_row);
LIGHTGBM_C_EXPORT int LGBM_DatasetPushRowsWithMetadata(
    DatasetHandle dataset, const void* data, int data_type, int32_t nrow, int32_t ncol,
    int32_t start_row, const float* labels, const float* weights,
    const double* init_scores, const int32_t* queries, int32_t tid);
LIGHTGBM_C_EXPORT int LGBM_DatasetCreateFromCSRFunc(void* get_row_funptr, int num_rows,
                                                    int64_t num_col, const char* parameters,
                                                    const DatasetHandle reference,
                                                    DatasetHandle* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetPushRowsByCSRWithMetadata(
    DatasetHandle dataset, const void* indptr, int indptr_type, const int32_t* indices,
    const void* data, int data_type, int64_t nindptr, int64_t nelem, int64_t start_row,
    const float* labels, const float* weights, const double* init_scores,
    const int32_t* queries, int32_t tid);
LIGHTGBM_C_EXPORT int LGBM_DatasetPushRowsByCSR(DatasetHandle dataset, const void* indptr,
                                                int indptr_type, const int32_t* indices,
                                                const void* data, int data_type,
                                                int64_t nindptr, int64_t nelem,
                                                int64_t num_col, int64_t start_row);
LIGHTGBM_C_EXPORT int LGBM_DatasetMarkFinished(DatasetHandle dataset);
LIGHTGBM_C_EXPORT int LGBM_DatasetSetWaitForManualFinish(DatasetHandle dataset, int wait);
typedef void* ByteBufferHandle;
/*! reference ABI: serialize the dataset schema into a library-owned ByteBuffer */
LIGHTGBM_C_EXPORT int LGBM_DatasetSerializeReferenceToBinary(DatasetHandle handle,
                                                             ByteBufferHandle* out,
                                                             int32_t* out_len);
LIGHTGBM_C_EXPORT int LGBM_ByteBufferGetAt(ByteBufferHandle handle, int32_t index,
                                           uint8_t* out_val);
LIGHTGBM_C_EXPORT int LGBM_ByteBufferFree(ByteBufferHandle handle);
LIGHTGBM_C_EXPORT int LGBM_DatasetCreateFromSerializedReference(
    const void* ref_buffer, int32_t ref_buffer_size, int64_t num_row,
    int32_t num_classes, const char* parameters, DatasetHandle* out);
struct ArrowArray;
struct ArrowSchema;
LIGHTGBM_C_EXPORT int LGBM_DatasetCreateFromArrow(int64_t n_chunks,
                                                  const struct ArrowArray* chunks,
                                                  const struct ArrowSchema* schema,
                                                  const char* parameters,
                                                  const DatasetHandle reference,
                                                  DatasetHandle* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetSetFieldFromArrow(DatasetHandle handle,
                                                    const char* field_name,
                                                    int64_t n_chunks,
                                                    const struct ArrowArray* chunks,
                                                    const struct ArrowSchema* schema);
LIGHTGBM_C_EXPORT int LGBM_DatasetGetSubset(const DatasetHandle handle,
                                            const int32_t* used_row_indices,
                                            int32_t num_used_row_indices,
                                            const char* parameters, DatasetHandle* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetSetFeatureNames(DatasetHandle handle,
                                                  const char** feature_names, int num);
LIGHTGBM_C_EXPORT int LGBM_DatasetGetFeatureNames(DatasetHandle handle, const int len,
                                                  int* num_feature_names,
                                                  const size_t buffer_len,
                                                  size_t* out_buffer_len, char** feature_names);
LIGHTGBM_C_EXPORT int LGBM_DatasetFree(DatasetHandle handle);
LIGHTGBM_C_EXPORT int LGBM_DatasetSaveBinary(DatasetHandle handle, const char* filename);
LIGHTGBM_C_EXPORT int LGBM_DatasetDumpText(DatasetHandle handle, const char* filename);
LIGHTGBM_C_EXPORT int LGBM_DatasetSetField(DatasetHandle handle, const char* field_name,
                                           const void* field_data, int num_element, int type);
LIGHTGBM_C_EXPORT int LGBM_DatasetGetField(DatasetHandle handle, const char* field_name,
                                           int* out_len, const void** out_ptr, int* out_type);
LIGHTGBM_C_EXPORT int LGBM_DatasetUpdateParamChecking(const char* old_parameters,
                                                      const char* new_parameters);
LIGHTGBM_C_EXPORT int LGBM_DatasetGetNumData(DatasetHandle handle, int32_t* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetGetNumFeature(DatasetHandle handle, int32_t* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetGetFeatureNumBin(DatasetHandle handle, int feature,
                                                   int32_t* out);
LIGHTGBM_C_EXPORT int LGBM_DatasetAddFeaturesFrom(DatasetHandle target, DatasetHandle source);

// ---- Booster
LIGHTGBM_C_EXPORT int LGBM_BoosterCreate(const DatasetHandle train_data,
                                         const char* parameters, BoosterHandle* out);
LIGHTGBM_C_EXPORT int LGBM_BoosterCreateFromModelfile(const char* filename,
                                                      int* out_num_iterations,
                                                      BoosterHandle* out);
LIGHTGBM_C_EXPORT int LGBM_BoosterLoadModelFromString(const char* model_str,
                                                      int* out_num_iterations,
                                                      BoosterHandle* out);
LIGHTGBM_C_EXPORT int LGBM_BoosterFree(BoosterHandle handle);
LIGHTGBM_C_EXPORT int LGBM_BoosterShuffleModels(BoosterHandle handle, int start_iter,
                                                int end_iter);
LIGHTGBM_C_EXPORT int LGBM_BoosterMerge(BoosterHandle handle,
                                        BoosterHandle other_handle);
LIGHTGBM_C_EXPORT int LGBM_BoosterAddValidData(BoosterHandle handle,
                                               const DatasetHandle valid_data);
LIGHTGBM_C_EXPORT int LGBM_BoosterResetTrainingData(BoosterHandle handle,
                                                    const DatasetHandle train_data);
LIGHTGBM_C_EXPORT int LGBM_BoosterResetParameter(BoosterHandle handle, const char* parameters);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetNumClasses(BoosterHandle handle, int* out_len);
LIGHTGBM_C_EXPORT int LGBM_BoosterUpdateOneIter(BoosterHandle handle, int* is_finished);
LIGHTGBM_C_EXPORT int LGBM_BoosterRefit(BoosterHandle handle, const int32_t* leaf_preds,
                                        int32_t nrow, int32_t ncol);
LIGHTGBM_C_EXPORT int LGBM_BoosterUpdateOneIterCustom(BoosterHandle handle, const float* grad,
                                                      const float* hess, int* is_finished);
LIGHTGBM_C_EXPORT int LGBM_BoosterRollbackOneIter(BoosterHandle handle);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetCurrentIteration(BoosterHandle handle,
                                                      int* out_iteration);
LIGHTGBM_C_EXPORT int LGBM_BoosterNumModelPerIteration(BoosterHandle handle,
                                                       int* out_tree_per_iteration);
LIGHTGBM_C_EXPORT int LGBM_BoosterNumberOfTotalModel(BoosterHandle handle, int* out_models);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetEvalCounts(BoosterHandle handle, int* out_len);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetEvalNames(BoosterHandle handle, const int len,
                                               int* out_len, const size_t buffer_len,
                                               size_t* out_buffer_len, char** out_strs);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetFeatureNames(BoosterHandle handle, const int len,
                                                  int* out_len, const size_t buffer_len,
                                                  size_t* out_buffer_len, char** out_strs);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetNumFeature(BoosterHandle handle, int* out_len);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetEval(BoosterHandle handle, int data_idx, int* out_len,
                                          double* out_results);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetNumPredict(BoosterHandle handle, int data_idx,
                                                int64_t* out_len);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetPredict(BoosterHandle handle, int data_idx,
                                             int64_t* out_len, double* out_result);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForFile(BoosterHandle handle,
                                                 const char* data_filename,
                                                 int data_has_header, int predict_type,
                                                 int start_iteration, int num_iteration,
                                                 const char* parameter,
                                                 const char* result_filename);
LIGHTGBM_C_EXPORT int LGBM_BoosterCalcNumPredict(BoosterHandle handle, int num_row,
                                                 int predict_type, int start_iteration,
                                                 int num_iteration, int64_t* out_len);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForCSR(BoosterHandle handle, const void* indptr,
                                                int indptr_type, const int32_t* indices,
                                                const void* data, int data_type,
                                                int64_t nindptr, int64_t nelem, int64_t num_col,
                                                int predict_type, int start_iteration,
                                                int num_iteration, const char* parameter,
                                                int64_t* out_len, double* out_result);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForCSRSingleRow(
    BoosterHandle handle, const void* indptr, int indptr_type, const int32_t* indices,
    const void* data, int data_type, int64_t nindptr, int64_t nelem, int64_t num_col,
    int predict_type, int start_iteration, int num_iteration, const char* parameter,
    int64_t* out_len, double* out_result);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForMat(BoosterHandle handle, const void* data,
                                                int data_type, int32_t nrow, int32_t ncol,
                                                int is_row_major, int predict_type,
                                                int start_iteration, int num_iteration,
                                                const char* parameter, int64_t* out_len,
                                                double* out_result);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForMatSingleRow(
    BoosterHandle handle, const void* data, int data_type, int ncol, int is_row_major,
    int predict_type, int start_iteration, int num_iteration, const char* parameter,
    int64_t* out_len, double* out_result);
/*! predict over an array of row pointers (reference c_api.h LGBM_BoosterPredictForMats) */
LIGHTGBM_C_EXPORT int LGBM_BoosterValidateFeatureNames(BoosterHandle handle,
                                                       const char** data_names,
                                                       int data_num_features);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetLoadedParam(BoosterHandle handle, int64_t buffer_len,
                                                 int64_t* out_len, char* out_str);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForCSC(BoosterHandle handle, const void* col_ptr,
                                                int col_ptr_type, const int32_t* indices,
                                                const void* data, int data_type,
                                                int64_t ncol_ptr, int64_t nelem,
                                                int64_t num_row, int predict_type,
                                                int start_iteration, int num_iteration,
                                                const char* parameter, int64_t* out_len,
                                                double* out_result);
/*! sparse (CSR) output of SHAP contributions; free with LGBM_BoosterFreePredictSparse */
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictSparseOutput(
    BoosterHandle handle, const void* indptr, int indptr_type, const int32_t* indices,
    const void* data, int data_type, int64_t nindptr, int64_t nelem, int64_t num_col,
    int predict_type, int start_iteration, int num_iteration, const char* parameter,
    int matrix_type, int64_t* out_len, void** out_indptr, int32_t** out_indices,
    void** out_data);
LIGHTGBM_C_EXPORT int LGBM_BoosterFreePredictSparse(void* indptr, int32_t* indices,
                                                    void* data, int indptr_type,
                                                    int data_type);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForArrow(BoosterHandle handle, int64_t n_chunks,
                                                  const struct ArrowArray* chunks,
                                                  const struct ArrowSchema* schema,
                                                  int predict_type, int start_iteration,
                                                  int num_iteration, const char* parameter,
                                                  int64_t* out_len, double* out_result);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForMats(BoosterHandle handle, const void** data,
                                                 int data_type, int32_t nrow, int32_t ncol,
                                                 int predict_type, int start_iteration,
                                                 int num_iteration, const char* parameter,
                                                 int64_t* out_len, double* out_result);
/*! single-row fast path: bind predict config once, then per-row calls with no re-parsing
 *  (reference c_api.h FastConfigHandle family) */
typedef void* FastConfigHandle;
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForMatSingleRowFastInit(
    BoosterHandle handle, int predict_type, int start_iteration, int num_iteration,
    int data_type, int32_t ncol, const char* parameter, FastConfigHandle* out_fastConfig);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForMatSingleRowFast(FastConfigHandle fastConfig_handle,
                                                             const void* data, int64_t* out_len,
                                                             double* out_result);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForCSRSingleRowFastInit(
    BoosterHandle handle, int predict_type, int start_iteration, int num_iteration,
    int data_type, int64_t num_col, const char* parameter, FastConfigHandle* out_fastConfig);
LIGHTGBM_C_EXPORT int LGBM_BoosterPredictForCSRSingleRowFast(
    FastConfigHandle fastConfig_handle, const void* indptr, int indptr_type,
    const int32_t* indices, const void* data, int64_t nindptr, int64_t nelem,
    int64_t* out_len, double* out_result);
LIGHTGBM_C_EXPORT int LGBM_FastConfigFree(FastConfigHandle fastConfig);
LIGHTGBM_C_EXPORT int LGBM_BoosterSaveModel(BoosterHandle handle, int start_iteration,
                                            int num_iteration, int feature_importance_type,
                                            const char* filename);
LIGHTGBM_C_EXPORT int LGBM_BoosterSaveModelToString(BoosterHandle handle, int start_iteration,
                                                    int num_iteration,
                                                    int feature_importance_type,
                                                    int64_t buffer_len, int64_t* out_len,
                                                    char* out_str);
LIGHTGBM_C_EXPORT int LGBM_BoosterDumpModel(BoosterHandle handle, int start_iteration,
                                            int num_iteration, int feature_importance_type,
                                            int64_t buffer_len, int64_t* out_len,
                                            char* out_str);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetLeafValue(BoosterHandle handle, int tree_idx,
                                               int leaf_idx, double* out_val);
LIGHTGBM_C_EXPORT int LGBM_BoosterSetLeafValue(BoosterHandle handle, int tree_idx,
                                               int leaf_idx, double val);
LIGHTGBM_C_EXPORT int LGBM_BoosterFeatureImportance(BoosterHandle handle, int num_iteration,
                                                    int importance_type, double* out_results);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetUpperBoundValue(BoosterHandle handle, double* out);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetLowerBoundValue(BoosterHandle handle, double* out);
LIGHTGBM_C_EXPORT int LGBM_BoosterGetLinear(BoosterHandle handle, int* out);

// ---- Network
typedef void (*AllgatherExtFunction)(char* input, int input_size, const int* block_start,
                                     const int* block_len, int num_block, char* output,
                                     int output_size);
typedef void (*ReduceScatterExtFunction)(char* input, int input_size, int type_size,
                                         const int* block_start, const int* block_len,
                                         int num_block, char* output, int output_size,
                                         const void* reducer);
LIGHTGBM_C_EXPORT int LGBM_NetworkInit(const char* machines, int local_listen_port,
                                       int listen_time_out, int num_machines);
LIGHTGBM_C_EXPORT int LGBM_NetworkFree();
LIGHTGBM_C_EXPORT int LGBM_NetworkInitWithFunctions(int num_machines, int rank,
                                                    void* reduce_scatter_ext_fun,
                                                    void* allgather_ext_fun);

#endif  // MIGBM_C_API_H_
