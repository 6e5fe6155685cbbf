// Minimal stand-in for brpc's butil/compiler_specific.h (see basictypes.h).
// The reference's gutil/port.h defines the PREDICT_* macros itself.
#pragma once
