/* gpu_dcompact_shim.h — test seam of the DB-side plugin.
 *
 * The shim calls the MI355X worker through this function-pointer table so
 * the translation test can substitute a recorder for the real libdcw.so
 * (which needs a gfx950 device).  Default resolution: dlopen("libdcw.so").
 */
#pragma once
#include "dcw.h"

#ifdef __cplusplus
extern "C" {
#endif

typedef struct dcw_shim_api {
  int32_t (*init)(int32_t device_ordinal);
  int32_t (*execute)(const dcw_job_desc*, dcw_job_result*);
  void (*free_result)(dcw_job_result*);
  void (*cancel)(int32_t job_id);
} dcw_shim_api;

void dcw_gpu_executor_set_api(const dcw_shim_api* api);

#ifdef __cplusplus
}
#endif
