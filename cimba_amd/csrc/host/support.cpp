// Host-side support: assert failure reporting (reference src/cmb_assert.c
// failure handler: reports context then aborts).
#include "cimba/config.hpp"

#include <stdio.h>
#include <stdlib.h>

namespace cmb {

[[noreturn]] void cmb_assert_fail_impl(const char* expr, const char* file, int line) {
    fprintf(stderr, "cimba_amd assertion failed: %s (%s:%d)\n", expr, file, line);
    fflush(stderr);
    abort();
}

}  // namespace cmb
