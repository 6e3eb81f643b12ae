// ORACLE/_REF — TEST INFRASTRUCTURE ONLY.
// grappolo's only mgp dependency is the per-thread allocation tracker
// (parallelLouvainMethod.cpp:186,229; buildNextPhase.cpp) — stub it out, as
// SURVEY.md §8c prescribes.
#include <mg_procedure.h>

extern "C" enum mgp_error mgp_track_current_thread_allocations(struct mgp_graph *) {
  return mgp_error::MGP_ERROR_NO_ERROR;
}
extern "C" enum mgp_error mgp_untrack_current_thread_allocations(struct mgp_graph *) {
  return mgp_error::MGP_ERROR_NO_ERROR;
}
