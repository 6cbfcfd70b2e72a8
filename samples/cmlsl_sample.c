/* The same plumbing check through the flat C API (reference analog:
 * tests/examples/mlsl_test/cmlsl_test.c exercising include/mlsl.h). */
#include <stdio.h>
#include <stdlib.h>

#include "mlsl/c_api.h"

#define COUNT 128

int main(void) {
    if (mlsl_init(-1, -1) != MLSL_SUCCESS) {
        fprintf(stderr, "init failed: %s\n", mlsl_last_error());
        return 1;
    }
    size_t rank = 0, size = 1;
    mlsl_rank(&rank);
    mlsl_world_size(&size);

    mlsl_distribution d;
    if (mlsl_distribution_create(size, 1, &d) != MLSL_SUCCESS) {
        fprintf(stderr, "dist failed: %s\n", mlsl_last_error());
        return 1;
    }

    float buf[COUNT];
    for (int i = 0; i < COUNT; ++i) buf[i] = (float)rank;

    mlsl_request req;
    if (mlsl_distribution_all_reduce(d, buf, buf, COUNT, MLSL_DT_F32, MLSL_RT_SUM,
                                     MLSL_GT_DATA, &req) != MLSL_SUCCESS) {
        fprintf(stderr, "allreduce failed: %s\n", mlsl_last_error());
        return 1;
    }
    void* result = NULL;
    if (mlsl_wait(req, &result) != MLSL_SUCCESS) {
        fprintf(stderr, "wait failed: %s\n", mlsl_last_error());
        return 1;
    }

    const float expected = (float)((size - 1) * size / 2.0);
    int bad = 0;
    for (int i = 0; i < COUNT; ++i)
        if (buf[i] != expected) ++bad;

    printf("[%zu/%zu] %s\n", rank, size, bad == 0 ? "PASSED" : "FAILED");
    mlsl_distribution_free(d);
    mlsl_finalize();
    return bad ? 1 : 0;
}
