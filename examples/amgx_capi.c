/* C client of the AMGX_* ABI (role of reference examples/amgx_capi.c:1):
 * read a MatrixMarket system, compose a solver from a JSON config file,
 * setup + solve, report status/iterations/residual.
 *
 *   ./amgx_capi -m examples/matrix.mtx -c configs/FGMRES_AGGREGATION.json \
 *               [-mode hDDI]
 *
 * Build: make -C csrc_capi example   (links ../csrc_capi/libamgx_amd.so)
 */
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

#include "amgx_c.h"

static void print_cb(const char *msg, int length)
{
    (void)length;
    printf("%s", msg);
}

static AMGX_Mode parse_mode(const char *s)
{
    if (!strcmp(s, "hDDI")) return AMGX_mode_hDDI;
    if (!strcmp(s, "hDFI")) return AMGX_mode_hDFI;
    if (!strcmp(s, "hFFI")) return AMGX_mode_hFFI;
    if (!strcmp(s, "dDDI")) return AMGX_mode_dDDI;
    if (!strcmp(s, "dDFI")) return AMGX_mode_dDFI;
    if (!strcmp(s, "dFFI")) return AMGX_mode_dFFI;
    fprintf(stderr, "unknown mode %s\n", s);
    exit(2);
}

int main(int argc, char **argv)
{
    const char *matrix_file = "examples/matrix.mtx";
    const char *config_file = "configs/FGMRES_AGGREGATION.json";
    AMGX_Mode mode = AMGX_mode_hDDI;
    int i;

    for (i = 1; i < argc; ++i) {
        if (!strcmp(argv[i], "-m") && i + 1 < argc) matrix_file = argv[++i];
        else if (!strcmp(argv[i], "-c") && i + 1 < argc)
            config_file = argv[++i];
        else if (!strcmp(argv[i], "-mode") && i + 1 < argc)
            mode = parse_mode(argv[++i]);
    }

    AMGX_SAFE_CALL(AMGX_initialize());
    AMGX_SAFE_CALL(AMGX_initialize_plugins());
    AMGX_SAFE_CALL(AMGX_register_print_callback(&print_cb));
    AMGX_SAFE_CALL(AMGX_install_signal_handler());

    {
        int major, minor;
        char *ver, *date, *time;
        AMGX_SAFE_CALL(AMGX_get_api_version(&major, &minor));
        AMGX_SAFE_CALL(AMGX_get_build_info_strings(&ver, &date, &time));
        printf("amgx_amd C ABI: api %d.%d, build %s (%s)\n", major, minor,
               ver, date);
    }

    AMGX_config_handle cfg;
    AMGX_SAFE_CALL(AMGX_config_create_from_file(&cfg, config_file));

    AMGX_resources_handle rsrc;
    AMGX_SAFE_CALL(AMGX_resources_create_simple(&rsrc, cfg));

    AMGX_matrix_handle A;
    AMGX_vector_handle b, x;
    AMGX_solver_handle solver;
    AMGX_SAFE_CALL(AMGX_matrix_create(&A, rsrc, mode));
    AMGX_SAFE_CALL(AMGX_vector_create(&b, rsrc, mode));
    AMGX_SAFE_CALL(AMGX_vector_create(&x, rsrc, mode));
    AMGX_SAFE_CALL(AMGX_solver_create(&solver, rsrc, mode, cfg));

    AMGX_SAFE_CALL(AMGX_read_system(A, b, x, matrix_file));

    int n, bx, by, nnz;
    AMGX_SAFE_CALL(AMGX_matrix_get_size(A, &n, &bx, &by));
    AMGX_SAFE_CALL(AMGX_matrix_get_nnz(A, &nnz));
    printf("system: n=%d nnz=%d block=%dx%d\n", n, nnz, bx, by);

    AMGX_SAFE_CALL(AMGX_vector_set_zero(x, n, bx));
    AMGX_SAFE_CALL(AMGX_solver_setup(solver, A));
    AMGX_SAFE_CALL(AMGX_solver_solve_with_0_initial_guess(solver, b, x));

    {
        AMGX_SOLVE_STATUS status;
        int iters;
        double resid;
        AMGX_SAFE_CALL(AMGX_solver_get_status(solver, &status));
        AMGX_SAFE_CALL(AMGX_solver_get_iterations_number(solver, &iters));
        AMGX_SAFE_CALL(AMGX_solver_get_iteration_residual(solver, -1, 0,
                                                          &resid));
        printf("status=%d iterations=%d final_residual=%.3e\n",
               (int)status, iters, resid);
        /* sanity: download the solution */
        double *sol = (double *)malloc((size_t)n * bx * sizeof(double));
        AMGX_SAFE_CALL(AMGX_vector_download(x, sol));
        printf("x[0]=%.6e x[n-1]=%.6e\n", sol[0], sol[(size_t)n * bx - 1]);
        free(sol);
        if (status != AMGX_SOLVE_SUCCESS) {
            fprintf(stderr, "solve did not converge\n");
            return 1;
        }
    }

    AMGX_SAFE_CALL(AMGX_solver_destroy(solver));
    AMGX_SAFE_CALL(AMGX_vector_destroy(x));
    AMGX_SAFE_CALL(AMGX_vector_destroy(b));
    AMGX_SAFE_CALL(AMGX_matrix_destroy(A));
    AMGX_SAFE_CALL(AMGX_resources_destroy(rsrc));
    AMGX_SAFE_CALL(AMGX_config_destroy(cfg));
    AMGX_SAFE_CALL(AMGX_finalize_plugins());
    AMGX_SAFE_CALL(AMGX_finalize());
    printf("AMGX_CAPI_OK\n");
    return 0;
}
