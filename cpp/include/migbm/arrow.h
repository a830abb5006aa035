/*!
 * migbm Arrow C data interface ingestion.
 * Capability parity target: reference include/LightGBM/arrow.h (+ LGBM_DatasetCreateFromArrow).
 * The struct layouts are the Arrow C ABI (stable, published at arrow.apache.org/docs/format/CDataInterface.html).
 */
#ifndef MIGBM_ARROW_H_
#define MIGBM_ARROW_H_

#include <cstdint>
#include <functional>
#include <string>
#include <vector>

extern "C" {

#ifndef ARROW_C_DATA_INTERFACE
#define ARROW_C_DATA_INTERFACE

#define ARROW_FLAG_DICTIONARY_ORDERED 1
#define ARROW_FLAG_NULLABLE 2
#define ARROW_FLAG_MAP_KEYS_SORTED 4

struct ArrowSchema {
  const char* format;
  const char* name;
  const char* metadata;
  int64_t flags;
  int64_t n_children;
  struct ArrowSchema** children;
  struct ArrowSchema* dictionary;
  void (*release)(struct ArrowSchema*);
  void* private_data;
};

struct ArrowArray {
  int64_t length;
  int64_t null_count;
  int64_t offset;
  int64_t n_buffers;
  int64_t n_children;
  const void** buffers;
  struct ArrowArray** children;
  struct ArrowArray* dictionary;
  void (*release)(struct ArrowArray*);
  void* private_data;
};

#endif  // ARROW_C_DATA_INTERFACE
}  // extern "C"

namespace migbm {

/*! value accessor for one primitive Arrow child array; returns NaN for nulls. */
std::function<double(int64_t)> ArrowColumnGetter(const ArrowArray* arr,
                                                 const ArrowSchema* schema);

}  // namespace migbm

#endif  // MIGBM_ARROW_H_
