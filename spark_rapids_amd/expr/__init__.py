from .expressions import CaseWhen, Expression, col, lit, when
from .aggregates import AggExpr, avg, count, count_star, max_, min_, sum_
from .windows import (WindowExpr, WindowFunc, dense_rank, lag, lead,
                      rank, row_number, win_avg, win_count, win_max,
                      win_min, win_sum)
