// trace.hip — roctx marker support (rocprofiler-sdk-roctx).
//
// The reference set SYCL's enable_profiling property but never read the
// event timestamps (reference concurency/main.cpp:145-146);
// here profiling is finished: hipEvent device times (conc.hip) plus roctx
// ranges that rocprofv3 --marker-trace picks up, so pattern phases are
// attributable in traces.

#include <rocprofiler-sdk-roctx/roctx.h>

namespace hpk {

void trace_push(const char* name) { roctxRangePush(name); }
void trace_pop() { roctxRangePop(); }
void trace_mark(const char* name) { roctxMarkA(name); }

} // namespace hpk
