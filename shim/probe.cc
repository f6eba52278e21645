#include "db/compaction/compaction_executor.h"
int main() { return 0; }
