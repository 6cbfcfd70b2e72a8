// Fatal-signal cleanup: on SEGV/BUS/ILL/ABRT/INT/TERM print a backtrace and
// tear down the comm context so peer ranks see clean EOFs instead of hangs.
// Reference analog: eplib/sig_handler.c:36-80 (client finalizes eplib + MPI
// then _exit(1)). Enabled by default; MLSL_HANDLE_SIGNALS=0 disables.
#pragma once

namespace mlsl {

void InstallSignalHandlers();
void RestoreSignalHandlers();

}  // namespace mlsl
