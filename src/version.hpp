#pragma once
#define RACON_MI355X_VERSION "v1.0.0-mi355x"
