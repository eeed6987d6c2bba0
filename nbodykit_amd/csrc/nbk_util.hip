// Library plumbing: version, error string, device probe.
#include "nbk_common.h"

char nbk_errbuf[1024] = "no error";

extern "C" const char* nbk_version(void) { return "nbk_hip 0.1.0 (gfx950)"; }

extern "C" const char* nbk_last_error_string(void) { return nbk_errbuf; }

extern "C" int nbk_device_count(void) {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}
