// ref_datetime_harness.cpp — golden-vector generator for the packed-DATETIME
// bit layout: compiles the REFERENCE's inline extraction functions
// (/root/reference/include/common/datetime.h: datetime_to_day:35,
// datetime_to_month:38, datetime_to_year:41, datetime_to_date:61,
// date_to_str:66) in place against our deterministic datetime generator, and
// emits JSON vectors committed as tests/golden/datetime_golden.json.
// Build: make -C oracle ref  (only where /root/reference is mounted).
#include "datetime.h"   // the reference header, via -I

#include "../include/bk_datagen.h"

int main() {
    printf("[\n");
    for (int i = 0; i < 256; i++) {
        uint64_t u = bk_mix64(0xDA7Eull + (uint64_t)i * 7919u);
        uint64_t dt = (uint64_t)bk_gen_datetime(u);
        uint32_t date = baikaldb::datetime_to_date(dt);
        std::string ds = baikaldb::date_to_str(date);
        printf("{\"dt\": %llu, \"year\": %u, \"month\": %u, \"day\": %u, "
               "\"date_str\": \"%s\"}%s\n",
               (unsigned long long)dt,
               baikaldb::datetime_to_year(dt),
               baikaldb::datetime_to_month(dt),
               baikaldb::datetime_to_day(dt),
               ds.c_str(), i + 1 < 256 ? "," : "");
    }
    printf("]\n");
    return 0;
}
